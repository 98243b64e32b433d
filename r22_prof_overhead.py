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


def run(steps, profiler=None):
    pipe = L7IngestPipeline(device="cuda", segment_rows=1 << 23,
                            dict_capacity=1 << 22,
                            time_base_s=cfg.base_time_ns // 10**9)
    pipe.segments.reserve(steps * 2_000_000 // (1 << 23) + 2)
    for _ in range(3):
        pipe.ingest_frame_payload(pay)          # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    if profiler is not None:
        prof = GpuProfiler(profiler)
        with prof.capture():
            for _ in range(steps):
                pipe.ingest_frame_payload(pay)
            torch.cuda.synchronize()
    else:
        for _ in range(steps):
            pipe.ingest_frame_payload(pay)
        torch.cuda.synchronize()
    return time.perf_counter() - t0


steps = 15
base = min(run(steps), run(steps))
profiled = run(steps, profiler=ProfilePipeline())
print({"steps": steps, "base_s": round(base, 3),
       "profiled_s": round(profiled, 3),
       "overhead_pct": round((profiled - base) / base * 100, 2)})
