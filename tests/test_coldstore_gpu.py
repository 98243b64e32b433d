"""GPU twins for the cold-segment codec: k_pack_bits/k_unpack_bits vs the
CPU reference packer (byte-identical words), and query-over-cold on the
GPU store."""
import pytest
import torch

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.store import coldstore as C

pytestmark = pytest.mark.gpu

CFG = SpanGenConfig(n=2000, seed=33, tag_cardinality=50, n_attrs=2,
                    n_ips=64, n_services=4, n_resources=12)


def test_pack_kernel_matches_cpu():
    torch.manual_seed(7)
    for bits in (1, 5, 11, 17, 23, 32):
        hi = (1 << bits) - 1
        vals = torch.randint(0, min(hi + 1, 2**31), (4097,),
                             dtype=torch.int64)
        v32 = vals.to(torch.int32)
        want = C.pack_stream(v32, 0, bits)                       # CPU
        got = C.pack_stream(v32.cuda(), 0, bits)                 # kernel
        torch.cuda.synchronize()
        assert torch.equal(got.cpu(), want), bits
        back = C.unpack_stream(got, 4097, 0, bits)
        torch.cuda.synchronize()
        assert torch.equal(back.cpu(), v32), bits


def test_query_over_cold_gpu():
    from deepflow_amd.ingest import L7IngestPipeline
    from deepflow_amd.query.engine import QueryEngine
    pipe = L7IngestPipeline(device="cuda", segment_rows=1 << 11,
                            dict_capacity=1 << 12,
                            time_base_s=CFG.base_time_ns // 10**9)
    pipe.ingest_frame_payload(gen_span_payload(CFG))
    pipe.ingest_frame_payload(gen_span_payload(CFG))
    eng = QueryEngine(pipe, device="cuda")
    queries = [
        "SELECT Count(*) AS c FROM l7_flow_log",
        "SELECT l7_protocol, Count(*) AS c, Avg(response_duration) AS a "
        "FROM l7_flow_log GROUP BY l7_protocol ORDER BY c DESC",
    ]
    want = [eng.query(q) for q in queries]
    seg = pipe.segments.segments[0]
    hot_bytes = seg.stored_bytes_per_row() * seg.n_rows
    assert pipe.segments.demote_oldest()
    cold_bytes = pipe.segments.cold[0].compressed_bytes()
    assert hot_bytes / cold_bytes > 1.5
    got = [eng.query(q) for q in queries]
    assert want == got
