"""Committed binary fixtures for the ingestion boundaries: the readers must
parse these EXACT bytes (generated once by the oracle writers, committed
under tests/golden/fixtures/) — guarding against writer+reader co-drift,
where a shared encoding mistake would hide in live roundtrips."""
import json
import os

import numpy as np
import pytest

from m3_amd import engine
from m3_amd.engine import FilesetVolume, CommitLog, parse_unaggregated

FIX = os.path.join(os.path.dirname(__file__), "golden", "fixtures")
START = 1427162400 * 10**9
pytestmark = pytest.mark.skipif(not engine.engine_available(),
                                reason="libm3gpu.so not built")


def test_fixture_fileset():
    exp = json.load(open(os.path.join(FIX, "fileset_expected.json")))
    with FilesetVolume(os.path.join(FIX, "fileset"), exp["block_start"]) as v:
        assert v.num_entries == len(exp["series"])
        by_id = {e[0].decode(): e for e in v.entries()}
        blob, offsets, lens = v.pack()
        import oracle
        o_off = np.concatenate([offsets, [np.uint64(len(blob))]])
        o_ts, o_vals, o_counts = oracle.decode_batch(blob, o_off, stride=64)
        order = [e[0].decode() for e in v.entries()]
        for s in exp["series"]:
            assert s["id"] in by_id
            i = order.index(s["id"])
            assert o_counts[i] == s["npts"]
            assert o_ts[i, :s["npts"]].tolist() == s["ts"]
            assert o_vals[i, :s["npts"]].tolist() == s["vals"]
            tags = by_id[s["id"]][4]
            assert tags == (s["tags"] or "").encode()


def test_fixture_commitlog():
    exp = json.load(open(os.path.join(FIX, "commitlog_expected.json")))
    with CommitLog(os.path.join(FIX, "commitlog-0-0.db")) as cl:
        assert cl.index == 3
        assert cl.num_series == 6
        for i, m in enumerate(cl.series()):
            e = exp[str(i)]
            assert m["id"] == f"fixture.cl.{i}".encode()
            assert m["ts"].tolist() == e["ts"]
            assert m["vals"].tolist() == e["vals"]


def test_fixture_unagg():
    out = parse_unaggregated(open(os.path.join(FIX, "unagg.bin"), "rb").read())
    assert [m["type"] for m in out] == [
        "counter", "batch_timer", "gauge", "timed_with_metadatas"]
    assert out[0]["counter_value"] == 12345
    assert out[0]["time_nanos"] == 777
    assert out[1]["values"].tolist() == [1.5, 2.25, 3.75]
    assert out[2]["values"][0] == -12.5
    assert out[3]["time_nanos"] == START
    assert out[3]["metric_type"] == 2
