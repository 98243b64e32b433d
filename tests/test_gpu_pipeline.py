"""GPU pipeline numerics tests: HIP kernels vs the CPU reference pipeline on
identical input. Requires an MI355X (run with -m gpu)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.ingest import L7IngestPipeline
from deepflow_amd.store import l7_schema as S
from deepflow_amd.store.kg import KnowledgeGraphTable, default_platform

N = 5000
# ip6_rate_pct > 0: the exact-equality tests below also cover the GPU
# ip6 decode path (pooled 16-byte addresses)
CFG = SpanGenConfig(n=N, seed=77, tag_cardinality=500, n_attrs=4,
                    n_ips=256, n_services=16, n_resources=100,
                    ip6_rate_pct=15)


def _mk(device):
    kg = KnowledgeGraphTable(capacity_pow2=1 << 12, device=device)
    kg.update(default_platform(CFG))
    p = L7IngestPipeline(device=device, segment_rows=1 << 14, kg=kg,
                        dict_capacity=1 << 14,
                        time_base_s=CFG.base_time_ns // 10**9)
    payload = gen_span_payload(CFG)
    p.ingest_frame_payload(payload)
    return p


@pytest.fixture(scope="module")
def pipes():
    assert torch.cuda.is_available(), "GPU required"
    from deepflow_amd.ops import native
    assert native.gpu().df_gpu_ready(), "HIP device not ready"
    gpu = _mk("cuda")
    torch.cuda.synchronize()
    cpu = _mk("cpu")
    return cpu, gpu


def test_fixed_columns_match(pipes):
    cpu, gpu = pipes
    a, b = cpu.segments.segments[0], gpu.segments.segments[0]
    n = N
    assert torch.equal(a.u64[:, :n], b.u64[:, :n].cpu())
    assert torch.equal(a.u32[:, :n], b.u32[:, :n].cpu())
    assert torch.equal(a.u8[:, :n], b.u8[:, :n].cpu())
    assert torch.equal(a.attr_cnt[:n], b.attr_cnt[:n].cpu())
    assert torch.equal(a.str_rowref[:n], b.str_rowref[:n].cpu())
    assert torch.equal(a.str_lens[:, :n], b.str_lens[:, :n].cpu())
    assert torch.equal(a.attr_start[:n], b.attr_start[:n].cpu())


def test_kg_join_matches(pipes):
    """GPU query-time KG join (in-kernel probe) == CPU oracle join."""
    from deepflow_amd.query.executor import _src_np, _kg_lookup_torch
    from deepflow_amd.query import spec as Q
    cpu, gpu = pipes
    a, b = cpu.segments.segments[0], gpu.segments.segments[0]
    rows = torch.arange(N, device="cuda")
    for idx in (0, S.N_KG - 1, S.N_KG, 2 * S.N_KG - 1):
        want = _src_np(a, Q.SRC_KG, idx, 0, 0, N, kg=cpu.kg)
        got = _kg_lookup_torch(gpu.kg, b, rows, idx).cpu().numpy()
        assert (want.astype("int64") == got).all(), idx
    # and through the query kernel: group by a KG tag
    from deepflow_amd.query.engine import QueryEngine
    qc = QueryEngine(cpu, device="cpu").query(
        "SELECT pod_id_1, Count(*) AS c FROM l7_flow_log "
        "GROUP BY pod_id_1 ORDER BY c DESC, pod_id_1 LIMIT 10")
    qg = QueryEngine(gpu, device="cuda").query(
        "SELECT pod_id_1, Count(*) AS c FROM l7_flow_log "
        "GROUP BY pod_id_1 ORDER BY c DESC, pod_id_1 LIMIT 10")
    assert qc == qg


def test_dict_hydration_matches(pipes):
    cpu, gpu = pipes
    a, b = cpu.segments.segments[0], gpu.segments.segments[0]
    for did_idx, (_, _, dom) in enumerate(S.DID_COLS):
        ha = cpu.dict.hydrate(dom, a.did[did_idx, :N].tolist())
        hb = gpu.dict.hydrate(dom, b.did[did_idx, :N].cpu().tolist())
        assert ha == hb
    # attrs
    sa = a.attr_start[:N].tolist()
    sb = b.attr_start[:N].cpu().tolist()
    ca = a.attr_cnt[:N].tolist()
    pa_pool = a.attr_pool.tolist()
    pb_pool = b.attr_pool.cpu().tolist()
    va = [pa_pool[sa[i] + ca[i]] if ca[i] else -1 for i in range(N)]
    vb = [pb_pool[sb[i] + ca[i]] if ca[i] else -1 for i in range(N)]
    ha = cpu.dict.hydrate(S.DICT_DOM_ATTR_VALUE, va)
    hb = gpu.dict.hydrate(S.DICT_DOM_ATTR_VALUE, vb)
    assert ha == hb


def test_dict_sizes_match(pipes):
    cpu, gpu = pipes
    assert cpu.dict.n_entries() == gpu.dict.n_entries()
    assert gpu.dict.dropped == 0


def test_pool_contents_match(pipes):
    cpu, gpu = pipes
    a, b = cpu.segments.segments[0], gpu.segments.segments[0]
    assert a.pool_len == b.pool_len
    tid = S.POOL_POS["trace_id"]
    pa = a.pool.numpy().tobytes()
    pb_ = b.pool.cpu().numpy().tobytes()
    for i in range(0, N, 97):
        oa = (int(a.str_rowref[i]) >> 16) + \
            sum(int(a.str_lens[c, i]) for c in range(tid))
        la = int(a.str_lens[tid, i])
        ob = (int(b.str_rowref[i]) >> 16) + \
            sum(int(b.str_lens[c, i]) for c in range(tid))
        lb = int(b.str_lens[tid, i])
        assert pa[oa:oa + la] == pb_[ob:ob + lb]


def test_metrics_match(pipes):
    cpu, gpu = pipes
    assert cpu.metrics.rows() == gpu.metrics.rows()
    for name in ("application.1s", "application.1m", "application_map.1s",
                 "application_map.1m"):
        ct, gt = cpu.rollups.get(name), gpu.rollups.get(name)
        assert ct.rows() == gt.rows(), name
        assert gt.drop_count() == 0, name


@pytest.mark.gpu
def test_checkpoint_roundtrip_gpu(tmp_path):
    """Checkpoint a GPU pipeline, restore into a fresh one, queries match."""
    from deepflow_amd.gen import SpanGenConfig
    from deepflow_amd.gen.spans import gen_span_payload
    from deepflow_amd.ingest import L7IngestPipeline
    from deepflow_amd.query.engine import QueryEngine
    from deepflow_amd.store import checkpoint as CK
    cfg = SpanGenConfig(n=5000, seed=77, tag_cardinality=64, n_attrs=2,
                        n_ips=64, n_services=4, n_resources=16)
    pipe = L7IngestPipeline(device="cuda", segment_rows=1 << 13,
                            dict_capacity=1 << 13,
                            time_base_s=cfg.base_time_ns // 10**9)
    pipe.ingest_frame_payload(gen_span_payload(cfg))
    eng = QueryEngine(pipe, device="cuda")
    q = ("SELECT request_resource, Count(*) AS c, Avg(response_duration) "
         "AS a FROM l7_flow_log GROUP BY request_resource ORDER BY c DESC")
    want = eng.query(q)
    path = str(tmp_path / "gpu.ckpt")
    CK.save_l7(pipe, path)
    fresh = L7IngestPipeline(device="cuda", segment_rows=1 << 13,
                             dict_capacity=1 << 13,
                             time_base_s=cfg.base_time_ns // 10**9)
    assert CK.load_l7(fresh, path) == cfg.n
    got = QueryEngine(fresh, device="cuda").query(q)
    assert want == got


@pytest.mark.gpu
def test_cold_checkpoint_gpu(tmp_path):
    """Cold (bit-packed) tier checkpoints and restores packed on GPU."""
    from deepflow_amd.gen import SpanGenConfig
    from deepflow_amd.gen.spans import gen_span_payload
    from deepflow_amd.ingest import L7IngestPipeline
    from deepflow_amd.query.engine import QueryEngine
    from deepflow_amd.store import checkpoint as CK
    cfg = SpanGenConfig(n=4000, seed=3, tag_cardinality=64, n_ips=64,
                        n_services=4, n_resources=8)
    p = L7IngestPipeline(device="cuda", segment_rows=1 << 12,
                         dict_capacity=1 << 13,
                         time_base_s=cfg.base_time_ns // 10**9)
    p.ingest_frame_payload(gen_span_payload(cfg))
    p.ingest_frame_payload(gen_span_payload(cfg))
    assert p.segments.demote_oldest()
    q = ("SELECT l7_protocol, Count(*) AS c FROM l7_flow_log "
         "GROUP BY l7_protocol")
    want = QueryEngine(p, device="cuda").query(q)
    path = str(tmp_path / "cold_gpu.ckpt")
    CK.save_l7(p, path)
    f = L7IngestPipeline(device="cuda", segment_rows=1 << 12,
                         dict_capacity=1 << 13,
                         time_base_s=cfg.base_time_ns // 10**9)
    assert CK.load_l7(f, path) == 8000
    assert len(f.segments.cold) == 1
    assert QueryEngine(f, device="cuda").query(q) == want
