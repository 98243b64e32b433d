"""Distributed query tests (gloo, world=2): cross-shard agg merge with
shard-local dictionary IDs hydrated pre-exchange; metric-bucket merge."""
import multiprocessing as mp
import os

import pytest


def _worker(rank, world, port, q):
    try:
        os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                          RANK=str(rank), WORLD_SIZE=str(world))
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)

        from deepflow_amd.gen import SpanGenConfig
        from deepflow_amd.gen.spans import gen_span_payload, gen_span_dict
        from deepflow_amd.ingest import L7IngestPipeline
        from deepflow_amd.query import QueryEngine
        from deepflow_amd.parallel.dist_query import (DistQueryEngine,
                                                      exchange_json,
                                                      merge_metric_rows)
        from deepflow_amd.store.kg import KnowledgeGraphTable, default_platform

        # same tag universe, disjoint spans per rank (seed differs)
        cfg = SpanGenConfig(n=150, seed=900 + rank, tag_cardinality=30,
                            n_services=6, n_ips=64, n_attrs=2)
        kg = KnowledgeGraphTable(capacity_pow2=1 << 10, device="cpu")
        kg.update(default_platform(cfg))
        pipe = L7IngestPipeline(device="cpu", segment_rows=1 << 9, kg=kg,
                               dict_capacity=1 << 12,
                               time_base_s=cfg.base_time_ns // 10**9)
        pipe.ingest_frame_payload(gen_span_payload(cfg))
        eng = DistQueryEngine(QueryEngine(pipe, device="cpu"), device="cpu")

        r1 = eng.query("SELECT Count(*) AS c FROM l7_flow_log")
        r2 = eng.query(
            "SELECT request_domain, Count(*) AS c, Avg(response_duration) "
            "AS a FROM l7_flow_log GROUP BY request_domain ORDER BY c DESC")
        # expected from both generators
        want = {}
        rrt = {}
        for rr in range(world):
            c2 = SpanGenConfig(n=150, seed=900 + rr, tag_cardinality=30,
                               n_services=6, n_ips=64, n_attrs=2)
            for i in range(150):
                t = gen_span_dict(c2, i)
                d = t["req"]["domain"]
                want[d] = want.get(d, 0) + 1
                rrt.setdefault(d, []).append(t["base"]["head"]["rrt"])
        ok_count = r1["values"] == [[300]]
        got = {row[0]: (row[1], row[2]) for row in r2["values"]}
        ok_groups = all(
            got[d][0] == want[d] and
            abs(got[d][1] - sum(rrt[d]) / len(rrt[d])) < 1e-6
            for d in want)
        # distributed percentile: histogram merge vs exact recompute
        rq = eng.query(
            "SELECT request_domain, Percentile(response_duration, 90) "
            "AS p90 FROM l7_flow_log GROUP BY request_domain")
        import numpy as np
        ok_pct = True
        got_p90 = {row[0]: row[1] for row in rq["values"]}
        for d, vals in rrt.items():
            exact = float(np.quantile(np.array(vals, dtype=np.float64),
                                      0.9))
            approx = got_p90[d]
            if abs(approx - exact) / max(exact, 1.0) > 0.08:
                ok_pct = False
        # distributed WITH: inner aggregate resolved globally, outer
        # re-aggregates the merged CTE rows identically on every rank
        rw = eng.query(
            "WITH per AS (SELECT request_domain, Count(*) AS c "
            "FROM l7_flow_log GROUP BY request_domain) "
            "SELECT Sum(c) AS total FROM per")
        ok_with = rw["values"] == [[300]]
        # topN pushdown: forced two-phase exchange must equal the full
        # exchange for a SLIMIT query
        sl = ("SELECT request_domain, Count(*) AS c FROM l7_flow_log "
              "GROUP BY request_domain SLIMIT 3")
        r_full = eng.query(sl)          # below threshold -> full exchange
        eng.pushdown_threshold = 0      # force candidate/delta phases
        r_push = eng.query(sl)
        eng.pushdown_threshold = DistQueryEngine.pushdown_threshold
        ok_topn = r_full == r_push and len(r_full["values"]) <= 3
        # metric bucket merge
        parts = exchange_json(pipe.metrics.rows(), "cpu")
        merged = merge_metric_rows(parts)
        total_req = sum(r["request"] for r in merged)
        dist.barrier()
        dist.destroy_process_group()
        q.put((rank, ok_count,
               ok_groups and ok_topn and ok_pct and ok_with, total_req))
    except Exception:
        import traceback
        q.put((rank, "ERR", traceback.format_exc(), None))


@pytest.mark.timeout(180)
def test_dist_query_two_ranks():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, 29517, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=170) for _ in range(2)]
    for p in procs:
        p.join(timeout=30)
    for r in results:
        assert r[1] != "ERR", r[2]
    for rank, ok_count, ok_groups, total_req in results:
        assert ok_count
        assert ok_groups
        assert total_req == 300
