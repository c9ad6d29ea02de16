"""Reference-shaped iteration over decoded batches.

The GPU engine returns decoded batches as SoA arrays; callers ported from
the reference often want the `encoding.ReaderIterator` shape
(dbnode/encoding/types.go:197-203: Next/Current/Err/Close) and the
`SeriesIterator` multi-iteration. These wrappers mirror those semantics
over decoded rows — host logic only, no compute (the GPU decode happened
at the batch boundary; see INTEGRATION.md §3 for the Go-side equivalent).

Semantics mirrored:
  - Next() advances and reports whether a point is available
    (iterator.go:81-100: first Next positions on the first point);
  - Current() returns (timestamp_ns, value, unit) for the current point
    and is only valid after a true Next() (types.go:200); CurrentAnnotation()
    gives the sticky annotation the reference returns as Current()'s third
    value (iterator.go:226-231) when the iterator was built with the
    annotation-set events of m3gpu_decode_batch_dev_ann;
  - Err() returns the sticky per-series error (the engine's
    M3GPU_SERIES_* code mapped to a string, or None);
  - Close() releases references (pooling is the caller's concern here).
"""
import numpy as np

from .engine import SERIES_ERRORS


class SliceReaderIterator:
    """`encoding.ReaderIterator` over one decoded series row."""

    def __init__(self, ts, vals, count, err=0, unit=1, ann_events=None):
        self._ts = ts
        self._vals = vals
        self._n = int(count)
        self._err = int(err)
        self._unit = unit
        self._i = -1
        # annotation-set events [(point_idx, bytes)] in stream order
        # (engine.parse_ann_region); carried forward like PrevAnt
        self._ann_events = ann_events or []
        self._ann_j = 0
        self._ann = None

    def Next(self):
        if self._err != 0:
            return False
        if self._i + 1 >= self._n:
            return False
        self._i += 1
        while (self._ann_j < len(self._ann_events) and
               self._ann_events[self._ann_j][0] <= self._i):
            self._ann = self._ann_events[self._ann_j][1]
            self._ann_j += 1
        return True

    def Current(self):
        if self._i < 0 or self._i >= self._n:
            raise RuntimeError("Current() before a successful Next()")
        return int(self._ts[self._i]), float(self._vals[self._i]), self._unit

    def CurrentAnnotation(self):
        """The reference Current()'s third return (sticky PrevAnt,
        iterator.go:226-231): bytes or None."""
        if self._i < 0 or self._i >= self._n:
            raise RuntimeError("CurrentAnnotation() before a successful Next()")
        return self._ann

    def Err(self):
        if self._err == 0:
            return None
        return SERIES_ERRORS.get(self._err, f"error {self._err}")

    def Close(self):
        self._ts = self._vals = None
        self._n = 0


class BatchIterators:
    """Per-series ReaderIterators over a decoded batch (SoA rows).

    Accepts numpy arrays or torch CPU tensors of shape [nseries, stride]
    plus per-series counts/errs, e.g. the (moved-to-host) outputs of
    `decode_batch_dev` or `fileset_ingest_dev`."""

    def __init__(self, ts, vals, counts, errs=None, unit=1, ann_regions=None):
        self._ts = np.asarray(ts)
        self._vals = np.asarray(vals)
        self._counts = np.asarray(counts)
        self._errs = np.asarray(errs) if errs is not None else None
        self._unit = unit
        # optional [nseries, ann_stride] uint8 regions from
        # m3gpu_decode_batch_dev_ann
        self._ann = np.asarray(ann_regions) if ann_regions is not None else None

    def __len__(self):
        return len(self._counts)

    def iterator(self, i):
        err = int(self._errs[i]) if self._errs is not None else 0
        events = None
        if self._ann is not None:
            from .engine import parse_ann_region
            events = parse_ann_region(self._ann[i])
        return SliceReaderIterator(self._ts[i], self._vals[i],
                                   self._counts[i], err, self._unit, events)

    def __iter__(self):
        for i in range(len(self)):
            yield self.iterator(i)
