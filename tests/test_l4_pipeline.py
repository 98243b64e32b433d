"""L4 (TaggedFlow) pipeline tests on CPU: decode oracle (generic pb decoder)
vs stored columns, KG join, network.1s rollup."""
import pytest

from deepflow_amd.gen import FlowGenConfig
from deepflow_amd.gen.flows import gen_flow_dict, gen_flow_payload
from deepflow_amd.ingest.l4_pipeline import L4IngestPipeline
from deepflow_amd.store import l4_schema as L4
from deepflow_amd.store import l7_schema as S
from deepflow_amd.store.kg import KnowledgeGraphTable, KgInfo

N = 150
CFG = FlowGenConfig(n=N, seed=31, n_ips=64, n_epcs=8)


@pytest.fixture(scope="module")
def pipe():
    kg = KnowledgeGraphTable(capacity_pow2=1 << 12, device="cpu")
    entries = {}
    for ipl in range(CFG.n_ips):
        ip = 0x0A000000 | ipl
        for epc in range(1, CFG.n_epcs + 1):
            entries[(epc, ip)] = KgInfo(pod_id=100 + ipl, subnet_id=1 + ipl % 4)
    kg.update(entries)
    p = L4IngestPipeline(device="cpu", segment_rows=1 << 9, kg=kg,
                        time_base_s=CFG.base_time_ns // 10**9)
    p.ingest_frame_payload(gen_flow_payload(CFG))
    return p


def test_rows(pipe):
    assert pipe.stats.flows_in == N


def test_columns(pipe):
    seg = pipe.segments.segments[0]
    for i in range(0, N, 11):
        t = gen_flow_dict(CFG, i)["flow"]
        assert int(seg.u64[L4.U64_COLS.index("byte_tx"), i]) == \
            t["metrics_peer_src"]["byte_count"]
        assert int(seg.u64[L4.U64_COLS.index("total_packet_rx"), i]) == \
            t["metrics_peer_dst"]["total_packet_count"]
        assert int(seg.u32[L4.U32_COLS.index("rtt"), i]) == \
            t["perf_stats"]["tcp"]["rtt"]
        assert int(seg.u32[L4.U32_COLS.index("retrans_tx"), i]) == \
            t["perf_stats"]["tcp"]["counts_peer_tx"].get("retrans_count", 0)
        assert int(seg.u8[L4.U8_COLS.index("close_type"), i]) == 1
        assert (int(seg.u32[L4.U32_COLS.index("ip4_0"), i]) & 0xFFFFFFFF) == \
            t["flow_key"]["ip_src"]


def test_kg(pipe):
    """Query-time KG join over L4 rows (no per-row kg block)."""
    from deepflow_amd.query.executor import _src_np
    from deepflow_amd.query import spec as Q
    seg = pipe.segments.segments[0]
    col = _src_np(seg, Q.SRC_KG, S.KG_COLS.index("pod_id"), 0, 0, N,
                  kg=pipe.kg)
    for i in range(0, N, 13):
        t = gen_flow_dict(CFG, i)["flow"]
        ip = t["flow_key"]["ip_src"]
        epc = t["metrics_peer_src"]["l3_epc_id"]
        info = pipe.kg.host.get((epc, ip))
        want = info.pod_id if info else 0
        assert int(col[i]) == want


def test_net1s(pipe):
    rows = pipe.metrics.rows()
    total_bytes = sum(r["byte_tx"] for r in rows)
    want = sum(gen_flow_dict(CFG, i)["flow"]["metrics_peer_src"]["byte_count"]
               for i in range(N))
    assert total_bytes == want
    assert sum(r["new_flow"] for r in rows) == 0  # generator sets no new_flow
    assert sum(r["closed_flow"] for r in rows) == N
