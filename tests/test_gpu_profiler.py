"""GPU profiler capture test (-m gpu): HIP kernels recorded via
torch.profiler (kineto/roctracer) land in the profile store."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from deepflow_amd.ingest.profile_pipeline import ProfilePipeline, build_flame
from deepflow_amd.profiler import GpuProfiler
from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.ingest import L7IngestPipeline
from deepflow_amd.store.kg import KnowledgeGraphTable, default_platform


def test_capture_ingest_kernels():
    assert torch.cuda.is_available()
    cfg = SpanGenConfig(n=5000, seed=3, tag_cardinality=100, n_ips=64)
    kg = KnowledgeGraphTable(capacity_pow2=1 << 12, device="cuda")
    kg.update(default_platform(cfg))
    ingest = L7IngestPipeline(device="cuda", segment_rows=1 << 14, kg=kg,
                             dict_capacity=1 << 14,
                             time_base_s=cfg.base_time_ns // 10**9)
    payload = gen_span_payload(cfg)
    profiles = ProfilePipeline()
    gp = GpuProfiler(profiles)
    with gp.capture():
        ingest.ingest_frame_payload(payload)
        torch.cuda.synchronize()
    st = profiles.store
    assert len(st.rows) > 0
    stacks = b"\n".join(st.id_to_loc).decode()
    assert "decode_l7" in stacks, stacks
    assert "intern" in stacks
    tree = build_flame(st.rows, st.id_to_loc,
                       event_type=GpuProfiler.EVENT_TYPE_ON_GPU)
    assert tree["value"] > 0
    gpu_node = tree["children"][0]
    assert gpu_node["name"] == "gpu"
