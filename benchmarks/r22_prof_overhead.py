"""Measure the continuous GPU profiler's overhead on the flagship ingest
step (reference claims <1% for its eBPF profiler; ours wraps the same
roctracer/kineto capture the flame-graph pipeline uses)."""
import time

import torch

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.ingest import L7IngestPipeline
from deepflow_amd.ingest.profile_pipeline import ProfilePipeline
from deepflow_amd.profiler.gpu_profiler import GpuProfiler

cfg = SpanGenConfig(n=2_000_000, seed=5, tag_cardinality=100_000,
                    n_attrs=4, n_ips=4096, n_services=64,
                    n_resources=2000)
pay = gen_span_payload(cfg)


from deepflow_amd.profiler.gpu_profiler import ContinuousGpuProfiler


def run(steps, mode="off"):
    pipe = L7IngestPipeline(device="cuda", segment_rows=1 << 23,
                            dict_capacity=1 << 22,
                            time_base_s=cfg.base_time_ns // 10**9)
    # provision like bench.py: no allocator or watermark work in the loop
    pipe.segments.reserve(steps * 2_000_000 // (1 << 23) + 2)
    for _ in range(3):
        pipe.ingest_frame_payload(pay)          # warmup
    torch.cuda.synchronize()
    prof = None
    if mode == "full":
        prof = GpuProfiler(ProfilePipeline())
    elif mode == "sampled":
        prof = ContinuousGpuProfiler(ProfilePipeline(), period=100)
    t0 = time.perf_counter()
    for _ in range(steps):
        if mode == "off":
            pipe.ingest_frame_payload(pay)
        elif mode == "full":
            with prof.capture():
                pipe.ingest_frame_payload(pay)
        else:
            with prof.step():
                pipe.ingest_frame_payload(pay)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return dt, prof.captures if prof else 0


steps = 100
base, _ = min(run(steps), run(steps))
sampled, caps = run(steps, mode="sampled")
full, _ = run(20, mode="full")
base20 = base / steps * 20
print({"steps": steps,
       "base_ms_per_step": round(base / steps * 1e3, 2),
       "sampled_ms_per_step": round(sampled / steps * 1e3, 2),
       "sampled_overhead_pct": round((sampled - base) / base * 100, 2),
       "captures": caps,
       "full_trace_overhead_pct": round((full - base20) / base20 * 100,
                                        1)})
