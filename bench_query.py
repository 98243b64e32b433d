#!/usr/bin/env python3
"""Query benchmark: DF-SQL group-by/filter over the in-HBM columnar store
(BASELINE config #4 shape, single GPU; the driver's bench.py covers ingest).

Ingests N synthetic spans once, then times representative queries; reports
rows scanned per second per query. Usage:
    python bench_query.py [--rows 30000000] [--iters 5]
"""
from __future__ import annotations

import argparse
import ctypes as ct
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.ingest import L7IngestPipeline
from deepflow_amd.ops import native
from deepflow_amd.query import QueryEngine
from deepflow_amd.store.kg import KnowledgeGraphTable, default_platform

QUERIES = [
    ("count_all", "SELECT Count(*) AS c FROM l7_flow_log"),
    ("group_domain",
     "SELECT request_domain, Count(*) AS c, Avg(response_duration) AS a "
     "FROM l7_flow_log GROUP BY request_domain"),
    ("group_resource",
     "SELECT request_resource, Count(*) AS c FROM l7_flow_log "
     "GROUP BY request_resource ORDER BY c DESC LIMIT 20"),
    ("timeseries_status",
     "SELECT time(60), response_status, Count(*) AS c FROM l7_flow_log "
     "GROUP BY time(60), response_status"),
    ("filtered_agg",
     "SELECT service_id_1, Sum(response_length) AS b FROM l7_flow_log "
     "WHERE response_status = 0 AND server_port = 8080 GROUP BY service_id_1"),
    ("point_select",
     "SELECT trace_id, request_domain FROM l7_flow_log "
     "WHERE flow_id = 1 LIMIT 10"),
]


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=30_000_000)
    ap.add_argument("--batch", type=int, default=4_000_000)
    ap.add_argument("--iters", type=int, default=5)
    ap.add_argument("--device", default=None)
    args = ap.parse_args()
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    if device == "cpu" and args.rows > 20000:
        args.rows, args.batch = 5000, 5000

    cfg = SpanGenConfig(n=args.batch, seed=1234, tag_cardinality=100_000,
                        n_ips=4096, n_services=256, n_resources=4096)
    kg = KnowledgeGraphTable(capacity_pow2=1 << 14, device=device)
    kg.update(default_platform(cfg))
    pipe = L7IngestPipeline(device=device, segment_rows=1 << 23, kg=kg,
                            dict_capacity=1 << 23,
                            time_base_s=cfg.base_time_ns // 10**9)
    pipe.segments.reserve(args.rows // (1 << 23) + 2)

    lib = native.cpu()
    c = native.span_cfg_c(cfg)
    done = 0
    t_load0 = time.perf_counter()
    while done < args.rows:
        n = min(args.batch, args.rows - done)
        need = int(lib.df_gen_spans_parallel(ct.byref(c), done, n, None, 0,
                                             None, None))
        buf = np.zeros(need, dtype=np.uint8)
        offs = np.zeros(n, dtype=np.uint32)
        lens = np.zeros(n, dtype=np.uint32)
        lib.df_gen_spans_parallel(ct.byref(c), done, n,
                                  buf.ctypes.data_as(ct.c_void_p), need,
                                  offs.ctypes.data_as(ct.c_void_p),
                                  lens.ctypes.data_as(ct.c_void_p))
        pipe.ingest(buf, offs, lens)
        done += n
    if device == "cuda":
        torch.cuda.synchronize()
    t_load1 = time.perf_counter()
    print(json.dumps({"loaded_rows": done,
                      "load_s": round(t_load1 - t_load0, 2),
                      "segments": len(pipe.segments.segments),
                      "device": device}))

    eng = QueryEngine(pipe, device=device)
    # the loaded store + dictionaries are long-lived: freeze them out of
    # the cyclic GC's gen2 scans (a gen2 pass over ~1M host objects
    # showed up as 60+ ms query-latency spikes)
    import gc
    gc.collect()
    gc.freeze()
    for name, sql in QUERIES:
        eng.query(sql)  # warmup (plan + allocs)
        if device == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        groups = 0
        for _ in range(args.iters):
            r = eng.query(sql)
            groups = len(r["values"])
        t1 = time.perf_counter()
        per_query = (t1 - t0) / args.iters
        print(json.dumps({
            "query": name,
            "ms": round(per_query * 1000, 3),
            "rows_scanned_per_s": round(done / per_query / 1e6, 1),
            "unit": "Mrows/s",
            "result_groups": groups,
        }))


if __name__ == "__main__":
    main()
