#!/usr/bin/env python3
"""BASELINE config #2: SmartEncoding ingest of a synthetic OTLP stream
(TracesData wire) on 1 MI355X with a 100k-cardinality tag dictionary.

The timed region covers the FULL OTLP path: native TracesData ->
AppProtoLogsData conversion (C++/OpenMP, ops/csrc/otlp_conv.cpp) +
record scan + H2D + the GPU decode/join/intern pipeline. OTLP-sourced
streams are conversion(CPU)-bound — the agent-native AppProtoLogsData
wire (bench.py) skips that cost, which is exactly why the reference
pushes conversion to its ingester fleet.
"""
import argparse
import json
import time

import numpy as np
import torch

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_dict
from deepflow_amd.ingest import L7IngestPipeline
from deepflow_amd.ops import native
from deepflow_amd.wire import pb, otlp


def build_otlp_batch(cfg: SpanGenConfig) -> bytes:
    def kv(k, v):
        return {"key": k, "value": {"string_value": v}}
    spans = []
    for i in range(cfg.n):
        t = gen_span_dict(cfg, i)
        attrs = [kv("http.method", t["req"]["req_type"] or "GET"),
                 kv("http.target", t["req"]["resource"]),
                 kv("http.host", t["req"]["domain"]),
                 {"key": "http.status_code",
                  "value": {"int_value": t["resp"].get("code", 200)}}]
        for nm, vv in zip(t["ext_info"].get("attribute_names", []),
                          t["ext_info"].get("attribute_values", [])):
            attrs.append(kv(nm, vv))
        ti = t.get("trace_info", {})
        spans.append({
            "trace_id": bytes.fromhex(ti["trace_id"])
            if ti.get("trace_id") else b"\x01" * 16,
            "span_id": bytes.fromhex(ti["span_id"]).ljust(8, b"\x00")[:8]
            if ti.get("span_id") else b"\x02" * 8,
            "name": t["req"]["resource"], "kind": 3,
            "start_time_unix_nano": t["base"]["start_time"],
            "end_time_unix_nano": t["base"]["end_time"],
            "attributes": attrs})
    return pb.encode({"resource_spans": [{
        "resource": {"attributes": [kv("service.name", "otlp-bench")]},
        "scope_spans": [{"spans": spans}]}]}, otlp.TRACES_DATA)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=500_000)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--device", default=None)
    args = ap.parse_args()
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    cfg = SpanGenConfig(n=args.batch, seed=11, tag_cardinality=100_000,
                        n_attrs=4, n_ips=4096, n_services=64,
                        n_resources=2000)
    t0 = time.time()
    blob = build_otlp_batch(cfg)
    print(json.dumps({"otlp_batch_spans": args.batch,
                      "otlp_bytes_per_span": round(len(blob) / args.batch,
                                                   1),
                      "gen_s": round(time.time() - t0, 1)}))
    lib = native.cpu()
    src = np.frombuffer(blob, dtype=np.uint8)
    need = int(lib.df_otlp_to_l7(src.ctypes.data, len(src), None, 0))
    dst = np.zeros(need, dtype=np.uint8)
    pipe = L7IngestPipeline(device=device, segment_rows=1 << 22,
                            dict_capacity=1 << 21,
                            time_base_s=cfg.base_time_ns // 10**9)
    total_rows = (args.steps + args.warmup) * args.batch
    pipe.segments.reserve(total_rows // (1 << 22) + 2)

    def step():
        lib.df_otlp_to_l7(src.ctypes.data, len(src), dst.ctypes.data, need)
        pipe.ingest_frame_payload(dst)

    for _ in range(args.warmup):
        step()
    if device == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if device == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    spans = args.steps * args.batch
    print(json.dumps({
        "metric": "otlp_spans_per_sec_ingested", "value":
        round(spans / dt, 1), "unit": "spans/s", "n_gpus": 1,
        "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": round(dt / args.steps * 1e3, 3),
        "higher_is_better": True, "data": "synthetic",
        "config": {"model": "otlp_traces_ingest (TracesData -> native "
                   "convert -> GPU pipeline)", "global_batch": args.batch,
                   "tag_cardinality": 100_000,
                   "dict_entries": pipe.dict.n_entries(),
                   "device": device}}))


if __name__ == "__main__":
    main()
