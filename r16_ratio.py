import torch, time
from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.ingest import L7IngestPipeline
cfg = SpanGenConfig(n=2000000, seed=5, tag_cardinality=100000, n_attrs=2,
                    n_ips=4096, n_services=64, n_resources=2000)
p = L7IngestPipeline(device="cuda", segment_rows=1 << 21, dict_capacity=1 << 21,
                     time_base_s=cfg.base_time_ns // 10**9)
pay = gen_span_payload(cfg)
p.ingest_frame_payload(pay)
p.ingest_frame_payload(pay)          # second segment so demote keeps a hot tail
seg = p.segments.segments[0]
hot = seg.stored_bytes_per_row() * seg.n_rows
torch.cuda.synchronize(); t0 = time.time()
assert p.segments.demote_oldest()
torch.cuda.synchronize(); t1 = time.time()
cold = p.segments.cold[0].compressed_bytes()
print({"rows": seg.n_rows, "hot_B_per_row": round(hot / 2e6, 1),
       "cold_B_per_row": round(cold / 2e6, 1), "ratio": round(hot / cold, 2),
       "compress_ms": round((t1 - t0) * 1e3, 1)})
t0 = time.time()
segs = p.segments.scan_list()
torch.cuda.synchronize(); t1 = time.time()
print({"materialize_ms": round((t1 - t0) * 1e3, 1), "segments": len(segs)})
from deepflow_amd.query.engine import QueryEngine
eng = QueryEngine(p, device="cuda")
t0 = time.time()
r = eng.query("SELECT l7_protocol, Count(*) AS c FROM l7_flow_log GROUP BY l7_protocol")
torch.cuda.synchronize(); t1 = time.time()
print({"groupby_over_cold_ms": round((t1 - t0) * 1e3, 1),
       "total_rows": sum(v[1] for v in r["values"])})
