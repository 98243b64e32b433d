"""Aux subsystem tests: throttler, 1m rollups, show tag values, eviction."""
import numpy as np

from deepflow_amd.utils.throttler import SamplingThrottler
from deepflow_amd.query.engine import rollup_rows
from deepflow_amd.store.segment import SegmentSet, L7Segment


def test_throttler_under_limit_passes_all():
    t = SamplingThrottler(limit_per_window=1000)
    keep = t.select(500, now_s=100)
    assert len(keep) == 500
    keep2 = t.select(400, now_s=100)
    assert len(keep2) == 400
    assert t.dropped == 0


def test_throttler_caps_window():
    t = SamplingThrottler(limit_per_window=100)
    total = 0
    for _ in range(10):
        total += len(t.select(50, now_s=7))
    assert total <= 100
    assert t.dropped == 500 - total
    # new window resets the budget but inherits last window's rate estimate
    # (500/s vs budget 100 -> p = 0.2), so ~20% of 50 pass
    kept = len(t.select(50, now_s=8))
    assert 1 <= kept <= 25


def test_throttler_deterministic():
    a = SamplingThrottler(limit_per_window=100, seed=1)
    b = SamplingThrottler(limit_per_window=100, seed=1)
    for w in (5, 5, 6):
        ka = a.select(300, now_s=w)
        kb = b.select(300, now_s=w)
        assert np.array_equal(ka, kb)


def test_rollup_rows():
    rows = [
        {"time": 100, "vtap_id": 1, "l7_protocol": 20, "response_status": 0,
         "server_port": 80, "request": 5, "rrt_max": 9, "rrt_sum": 50},
        {"time": 110, "vtap_id": 1, "l7_protocol": 20, "response_status": 0,
         "server_port": 80, "request": 7, "rrt_max": 20, "rrt_sum": 70},
        {"time": 111, "vtap_id": 2, "l7_protocol": 20, "response_status": 0,
         "server_port": 80, "request": 1, "rrt_max": 1, "rrt_sum": 1},
    ]
    out = rollup_rows(rows, 60)
    assert len(out) == 2
    r1 = next(r for r in out if r["vtap_id"] == 1)
    assert r1["time"] == 60 and r1["request"] == 12
    assert r1["rrt_max"] == 20 and r1["rrt_sum"] == 120


def test_segment_eviction():
    ss = SegmentSet(segment_rows=64, device="cpu", cls=L7Segment,
                    max_bytes=SegmentSet.seg_alloc_bytes(
                        L7Segment(64, "cpu")) * 2 + 1000)
    for _ in range(5):
        seg = ss.tail(64)
        seg.n_rows = 64
    assert len(ss.segments) <= 3
    assert ss.evicted_segments >= 2
    assert ss.evicted_rows == ss.evicted_segments * 64
