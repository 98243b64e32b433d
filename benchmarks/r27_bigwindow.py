"""Big-window stress: fill a large hot window, demote half to the
bit-packed cold tier, query across the whole history (hot + cold).
Exercises the 288 GB sizing story end to end on one GPU."""
import json
import time

import torch

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.ingest import L7IngestPipeline
from deepflow_amd.query.engine import QueryEngine

BATCH = 4_000_000
STEPS = 25  # 100M spans total

cfg = SpanGenConfig(n=BATCH, seed=5, tag_cardinality=100_000, n_attrs=2,
                    n_ips=4096, n_services=64, n_resources=2000)
pipe = L7IngestPipeline(device="cuda", segment_rows=1 << 22,
                        dict_capacity=1 << 22,
                        time_base_s=cfg.base_time_ns // 10**9)
pay = gen_span_payload(cfg)
t0 = time.time()
for i in range(STEPS):
    pipe.ingest_frame_payload(pay)
torch.cuda.synchronize()
load_s = time.time() - t0
hot_bytes = pipe.segments.total_alloc_bytes()
print(json.dumps({"rows": pipe.segments.n_rows, "load_s": round(load_s, 1),
                  "hot_gb": round(hot_bytes / 2**30, 1),
                  "segments": len(pipe.segments.segments)}))
# demote the older half
t0 = time.time()
target = len(pipe.segments.segments) // 2
for _ in range(target):
    pipe.segments.demote_oldest()
torch.cuda.synchronize()
cold_bytes = sum(c.compressed_bytes() for c in pipe.segments.cold)
print(json.dumps({"demoted": target, "demote_s": round(time.time() - t0, 1),
                  "cold_gb": round(cold_bytes / 2**30, 2),
                  "cold_rows": sum(c.n_rows for c in pipe.segments.cold),
                  "free_list": len(pipe.segments._free)}))
eng = QueryEngine(pipe, device="cuda")
for name, q in [
    ("count_all", "SELECT Count(*) AS c FROM l7_flow_log"),
    ("group_status", "SELECT response_status, Count(*) AS c, "
     "Avg(response_duration) AS a FROM l7_flow_log "
     "GROUP BY response_status"),
]:
    t0 = time.time()
    r = eng.query(q)
    torch.cuda.synchronize()
    print(json.dumps({"query": name, "ms": round((time.time() - t0) * 1e3,
                                                 1),
                      "rows_covered": pipe.segments.n_rows,
                      "result_head": r["values"][:3]}))
