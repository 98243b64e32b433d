"""Two-tier soak: ingest with inline demotion (hot watermark forces the
oldest segment to compress while writes continue) — the production
steady state of the 288 GB window."""
import json
import time

import torch

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.ingest import L7IngestPipeline

cfg = SpanGenConfig(n=2_000_000, seed=5, tag_cardinality=100_000,
                    n_attrs=2, n_ips=4096, n_services=64, n_resources=2000)
pipe = L7IngestPipeline(device="cuda", segment_rows=1 << 22,
                        dict_capacity=1 << 22,
                        time_base_s=cfg.base_time_ns // 10**9)
pipe.segments.hot_max_bytes = 8 << 30     # hot window ~4 segments
pipe.segments.max_bytes = 40 << 30        # cold tier cap
pay = gen_span_payload(cfg)
for _ in range(3):
    pipe.ingest_frame_payload(pay)
torch.cuda.synchronize()
t0 = time.perf_counter()
STEPS = 40
for _ in range(STEPS):
    pipe.ingest_frame_payload(pay)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(json.dumps({
    "spans_per_sec_with_inline_demotion": round(STEPS * cfg.n / dt, 0),
    "ms_per_step": round(dt / STEPS * 1e3, 2),
    "hot_segments": len(pipe.segments.segments),
    "cold_segments": len(pipe.segments.cold),
    "cold_rows": sum(c.n_rows for c in pipe.segments.cold),
    "evicted_rows": pipe.segments.evicted_rows}))
from deepflow_amd.query.engine import QueryEngine
eng = QueryEngine(pipe, device="cuda")
t0 = time.perf_counter()
r = eng.query("SELECT response_status, Count(*) AS c FROM l7_flow_log "
              "GROUP BY response_status")
torch.cuda.synchronize()
print(json.dumps({"cross_tier_query_ms":
                  round((time.perf_counter() - t0) * 1e3, 1),
                  "rows": pipe.segments.n_rows,
                  "head": r["values"][:2]}))
