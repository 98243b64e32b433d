"""Profile pipeline + flame graph tests."""
import pytest
from fastapi.testclient import TestClient

from deepflow_amd.ingest.profile_pipeline import (ProfilePipeline, build_flame)
from deepflow_amd.server import DeepflowServer
from deepflow_amd.wire import pb, metric, framing


def mk_profile(stacks, process="svc-a", event_type=1, ts=1000000):
    data = "\n".join(f"{s} {c}" for s, c in stacks).encode()
    return {
        "name": "app", "units": "samples", "format": "folded",
        "data": data, "timestamp": ts, "event_type": event_type,
        "pid": 42, "process_name": process, "spy_name": "ebpf",
        "count": 1,
    }


def test_folded_ingest_and_flame():
    pipe = ProfilePipeline()
    p1 = mk_profile([("main;work;io_read", 5), ("main;work;cpu_spin", 3),
                     ("main;idle", 2)])
    p2 = mk_profile([("main;work;io_read", 4)])
    payload = framing.pack_records([
        pb.encode(p1, metric.PROFILE), pb.encode(p2, metric.PROFILE)])
    assert pipe.ingest_payload(payload) == 2
    st = pipe.store
    assert len(st.id_to_loc) == 3  # dedup across profiles
    tree = build_flame(st.rows, st.id_to_loc)
    assert tree["value"] == 14
    main = tree["children"][0]
    assert main["name"] == "main" and main["value"] == 14
    work = next(c for c in main["children"] if c["name"] == "work")
    assert work["value"] == 12
    io = next(c for c in work["children"] if c["name"] == "io_read")
    assert io["value"] == 9 and io["self"] == 9


def test_profile_http():
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 9,
                         dict_capacity=1 << 10)
    payload = framing.pack_records([
        pb.encode(mk_profile([("a;b", 7)], process="px"), metric.PROFILE)])
    frame = framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_PROFILE), payload)
    assert srv.receiver.handle_frame(frame)
    client = TestClient(srv.app)
    assert client.get("/v1/profile/processes").json() == ["px"]
    tree = client.get("/v1/profile/flame",
                      params={"process_name": "px"}).json()
    assert tree["value"] == 7


def test_zstd_compressed_data():
    import ctypes as ct
    import numpy as np
    from deepflow_amd.ops import native
    lib = native.cpu()
    raw = b"x;y;z 11"
    src = np.frombuffer(raw, dtype=np.uint8)
    dst = np.zeros(1024, dtype=np.uint8)
    n = lib.df_zstd_compress(src.ctypes.data, len(raw), dst.ctypes.data,
                             1024, 3)
    d = mk_profile([], ts=5)
    d["data"] = dst[:n].tobytes()
    d["data_compressed"] = 1
    pipe = ProfilePipeline()
    pipe.ingest_profile(d)
    assert pipe.store.id_to_loc == [b"x;y;z"]
    assert pipe.store.rows[0].value == 11


def test_continuous_profiler_duty_cycle():
    """period/window duty logic: captures only its share of steps."""
    import contextlib
    from deepflow_amd.profiler.gpu_profiler import ContinuousGpuProfiler

    class Probe(ContinuousGpuProfiler):
        @contextlib.contextmanager
        def capture(self):
            self.captures += 1
            yield None

    prof = Probe(pipeline=None, period=5, window=2, interval_s=0.0)
    ran = 0
    for _ in range(20):
        with prof.step():
            ran += 1
    assert ran == 20
    assert prof.captures == 8  # 2 of every 5 steps
