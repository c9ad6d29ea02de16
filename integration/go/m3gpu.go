// Package m3gpu binds the MI355X-native M3TSZ + rollup engine
// (libm3gpu.so, C ABI in include/m3gpu.h) for use inside m3db/m3.
//
// This file is the drop-in surface a maintainer vendors into the reference:
// it mirrors the reference's pluggable codec boundary (alloc functions
// registered into pools at dbnode/storage/options.go:498-510,
// dbnode/server/server.go:1786-1793, dbnode/client/options.go:504) for the
// BULK paths — block ingestion, cold-flush merges, AggregateTiles, bulk
// fetches — while single-point streaming stays on the existing Go codec.
// See INTEGRATION.md for the full wiring discussion.
//
// NOTE: this tree's build environment has no Go toolchain (SURVEY.md §0);
// this file is compile-checked only where Go is available. It contains no
// logic beyond argument marshalling — all semantics live behind the C ABI
// and are pinned by the oracle + GPU parity suites.
package m3gpu

/*
#cgo CFLAGS: -I${SRCDIR}/../../include
#cgo LDFLAGS: -L${SRCDIR}/../../m3_amd/csrc -lm3gpu
#include <stdlib.h>
#include "m3gpu.h"
*/
import "C"

import (
	"encoding/binary"
	"errors"
	"unsafe"
)

// SeriesError mirrors the per-series M3GPU_SERIES_* codes.
type SeriesError int32

const (
	SeriesOK             SeriesError = 0
	SeriesEOF            SeriesError = 1
	SeriesDodOverflow    SeriesError = 2
	SeriesNoScheme       SeriesError = 3
	SeriesInvalidMult    SeriesError = 4
	SeriesAnnotation     SeriesError = 5
	SeriesCapacity       SeriesError = 6
	SeriesUnsorted       SeriesError = 7
	SeriesBucketOverflow SeriesError = 8
)

func lastError() error {
	return errors.New(C.GoString(C.m3gpu_last_error()))
}

// Init selects the GPU device for this process (one engine per GPU; shard
// the series space across devices exactly as m3 shards across instances).
func Init(device int) error {
	if rc := C.m3gpu_init(C.int(device)); rc != 0 {
		return lastError()
	}
	return nil
}

// DecodeBatch decodes packed M3TSZ streams (layout contract in m3gpu.h:
// offsets 16-byte aligned, zero padded) into SoA rows of `stride` points.
// Replaces a loop of m3tsz.NewReaderIterator/Next/Current
// (dbnode/encoding/m3tsz/iterator.go:81-219) over a batch of blocks.
func DecodeBatch(
	blobs []byte, offsets []uint64, lens []uint32,
	intOptimized bool, defaultUnit byte, stride uint32,
) (ts []int64, vals []float64, counts []uint32, errs []SeriesError, err error) {
	n := uint32(len(lens))
	if n == 0 {
		return nil, nil, nil, nil, nil
	}
	ts = make([]int64, uint64(n)*uint64(stride))
	vals = make([]float64, uint64(n)*uint64(stride))
	counts = make([]uint32, n)
	errs = make([]SeriesError, n)
	intOpt := C.int(0)
	if intOptimized {
		intOpt = 1
	}
	rc := C.m3gpu_decode_batch(
		(*C.uint8_t)(unsafe.Pointer(&blobs[0])), C.uint64_t(len(blobs)),
		(*C.uint64_t)(unsafe.Pointer(&offsets[0])),
		(*C.uint32_t)(unsafe.Pointer(&lens[0])),
		C.uint32_t(n), intOpt, C.uint8_t(defaultUnit),
		(*C.int64_t)(unsafe.Pointer(&ts[0])),
		(*C.double)(unsafe.Pointer(&vals[0])),
		(*C.uint32_t)(unsafe.Pointer(&counts[0])),
		(*C.int32_t)(unsafe.Pointer(&errs[0])), C.uint32_t(stride))
	if rc != 0 {
		return nil, nil, nil, nil, lastError()
	}
	return ts, vals, counts, errs, nil
}

// EncodeBatch encodes SoA rows into finalized M3TSZ streams (EOS tail
// included), byte-identical to m3tsz.Encoder output (encoder.go:89-250).
// outStride must be a multiple of 8 and hold the worst case (~24 B/pt + 32).
func EncodeBatch(
	ts []int64, vals []float64, counts []uint32,
	stride uint32, intOptimized bool, unit byte, outStride uint32,
) (streams []byte, lens []uint32, errs []SeriesError, err error) {
	n := uint32(len(counts))
	if n == 0 {
		return nil, nil, nil, nil
	}
	streams = make([]byte, uint64(n)*uint64(outStride))
	lens = make([]uint32, n)
	errs = make([]SeriesError, n)
	intOpt := C.int(0)
	if intOptimized {
		intOpt = 1
	}
	rc := C.m3gpu_encode_batch(
		(*C.int64_t)(unsafe.Pointer(&ts[0])),
		(*C.double)(unsafe.Pointer(&vals[0])),
		(*C.uint32_t)(unsafe.Pointer(&counts[0])),
		C.uint32_t(n), C.uint32_t(stride), intOpt, C.uint8_t(unit),
		(*C.uint8_t)(unsafe.Pointer(&streams[0])), C.uint32_t(outStride),
		(*C.uint32_t)(unsafe.Pointer(&lens[0])),
		(*C.int32_t)(unsafe.Pointer(&errs[0])))
	if rc != 0 {
		return nil, nil, nil, nil, lastError()
	}
	return streams, lens, errs, nil
}

// RollupBatch runs the fused decode -> windowed rollup with m3aggregator
// semantics (generic_elem.go AddValue/Consume + aggregation/{counter,gauge,
// timer}.go). aggTypes carries the reference's aggregation.Type values
// (metrics/aggregation/type.go:31-56). Output timestamps are window ENDS
// (list.go:541-543), so flush handlers consume them unchanged; this is the
// backend AggregateTiles (storage/shard.go:2682) drives.
func RollupBatch(
	blobs []byte, offsets []uint64, lens []uint32,
	intOptimized bool, defaultUnit byte,
	metricType int, windowNs int64, nbuckets uint32,
	aggTypes []int32,
) (out []float64, windowTs []int64, errs []SeriesError, err error) {
	n := uint32(len(lens))
	if n == 0 || len(aggTypes) == 0 {
		return nil, nil, nil, nil
	}
	out = make([]float64, uint64(n)*uint64(nbuckets)*uint64(len(aggTypes)))
	windowTs = make([]int64, uint64(n)*uint64(nbuckets))
	errs = make([]SeriesError, n)
	intOpt := C.int(0)
	if intOptimized {
		intOpt = 1
	}
	rc := C.m3gpu_rollup_batch(
		(*C.uint8_t)(unsafe.Pointer(&blobs[0])), C.uint64_t(len(blobs)),
		(*C.uint64_t)(unsafe.Pointer(&offsets[0])),
		(*C.uint32_t)(unsafe.Pointer(&lens[0])),
		C.uint32_t(n), intOpt, C.uint8_t(defaultUnit),
		C.int(metricType), C.int64_t(windowNs), C.uint32_t(nbuckets),
		(*C.int32_t)(unsafe.Pointer(&aggTypes[0])), C.int(len(aggTypes)),
		(*C.double)(unsafe.Pointer(&out[0])),
		(*C.int64_t)(unsafe.Pointer(&windowTs[0])),
		(*C.int32_t)(unsafe.Pointer(&errs[0])))
	if rc != 0 {
		return nil, nil, nil, nil, lastError()
	}
	return out, windowTs, errs, nil
}

// AnnotationEvent is one annotation-set event from the annotation-capturing
// decode: the annotation starts applying at Point (0-based) and stays the
// Current() third return (iterator.go:226-231 sticky PrevAnt) until the
// next event.
type AnnotationEvent struct {
	Point uint32
	Bytes []byte
}

// DecodeBatchAnn decodes like DecodeBatch and additionally captures every
// annotation-set event per series (m3gpu_decode_batch_ann; region layout in
// m3gpu.h). annStride sizes the per-series region (4-aligned, >= 16); a
// series whose annotations overflow it flags SeriesCapacity but still
// decodes its values.
func DecodeBatchAnn(
	blobs []byte, offsets []uint64, lens []uint32,
	intOptimized bool, defaultUnit byte, stride uint32, annStride uint32,
) (ts []int64, vals []float64, counts []uint32, errs []SeriesError,
	anns [][]AnnotationEvent, err error) {
	n := uint32(len(lens))
	if n == 0 {
		return nil, nil, nil, nil, nil, nil
	}
	ts = make([]int64, uint64(n)*uint64(stride))
	vals = make([]float64, uint64(n)*uint64(stride))
	counts = make([]uint32, n)
	errs = make([]SeriesError, n)
	region := make([]byte, uint64(n)*uint64(annStride))
	intOpt := C.int(0)
	if intOptimized {
		intOpt = 1
	}
	rc := C.m3gpu_decode_batch_ann(
		(*C.uint8_t)(unsafe.Pointer(&blobs[0])), C.uint64_t(len(blobs)),
		(*C.uint64_t)(unsafe.Pointer(&offsets[0])),
		(*C.uint32_t)(unsafe.Pointer(&lens[0])),
		C.uint32_t(n), intOpt, C.uint8_t(defaultUnit),
		(*C.int64_t)(unsafe.Pointer(&ts[0])),
		(*C.double)(unsafe.Pointer(&vals[0])),
		(*C.uint32_t)(unsafe.Pointer(&counts[0])),
		(*C.int32_t)(unsafe.Pointer(&errs[0])), C.uint32_t(stride),
		(*C.uint8_t)(unsafe.Pointer(&region[0])), C.uint32_t(annStride))
	if rc != 0 {
		return nil, nil, nil, nil, nil, lastError()
	}
	anns = make([][]AnnotationEvent, n)
	le := binary.LittleEndian
	for i := uint32(0); i < n; i++ {
		r := region[uint64(i)*uint64(annStride) : uint64(i+1)*uint64(annStride)]
		nev := le.Uint32(r[0:4])
		evs := make([]AnnotationEvent, 0, nev)
		for j := uint32(0); j < nev; j++ {
			ev := r[4+j*12 : 4+j*12+12]
			off, ln := le.Uint32(ev[4:8]), le.Uint32(ev[8:12])
			evs = append(evs, AnnotationEvent{
				Point: le.Uint32(ev[0:4]),
				Bytes: append([]byte(nil), r[off:off+ln]...),
			})
		}
		anns[i] = evs
	}
	return ts, vals, counts, errs, anns, nil
}
