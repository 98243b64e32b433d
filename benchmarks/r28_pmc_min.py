"""Minimal kernel exerciser for PMC counter collection."""
import torch
from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.ingest import L7IngestPipeline

cfg = SpanGenConfig(n=500_000, seed=5, tag_cardinality=100_000, n_attrs=2,
                    n_ips=4096, n_services=64, n_resources=2000)
pipe = L7IngestPipeline(device="cuda", segment_rows=1 << 21,
                        dict_capacity=1 << 21,
                        time_base_s=cfg.base_time_ns // 10**9)
pay = gen_span_payload(cfg)
for _ in range(2):
    pipe.ingest_frame_payload(pay)
torch.cuda.synchronize()
print("done", pipe.segments.n_rows)
