"""Distributed query execution across GPU shards.

Span/flow streams are hash-sharded by agent across ranks (one rank per
GPU); queries run shard-local and merge on collectives (RCCL over xGMI on
cuda, gloo on cpu) — the reference's "each analyzer independent, merge at
query time" model with the scatter of ClickHouse sub-queries
replaced by an all-gather of partials.

Aggregate queries exchange (hydrated key tuple, RAW agg vector): dict IDs
are shard-local so keys hydrate before the exchange, while avg stays split
as (sum, count) so cross-shard merging is exact. 1s rollup tables merge the
same way via merge_metric_rows (the time-bucket reduce step).
"""
from __future__ import annotations

import json
from typing import Dict, List

import torch
import torch.distributed as dist

from ..query import spec as Q


def _gather_variable_u8(t: torch.Tensor, device: str) -> List[torch.Tensor]:
    n = torch.tensor([t.numel()], dtype=torch.int64, device=device)
    world = dist.get_world_size()
    sizes = [torch.zeros(1, dtype=torch.int64, device=device)
             for _ in range(world)]
    dist.all_gather(sizes, n)
    max_n = max(int(s.item()) for s in sizes)
    pad = torch.zeros(max_n, dtype=torch.uint8, device=device)
    if t.numel():
        pad[: t.numel()] = t
    out = [torch.zeros(max_n, dtype=torch.uint8, device=device)
           for _ in range(world)]
    dist.all_gather(out, pad)
    return [o[: int(s.item())] for o, s in zip(out, sizes)]


def exchange_json(obj, device: str = "cpu") -> List:
    """all_gather a JSON-serializable object from every rank (byte-tensor
    transport: works on gloo and nccl/RCCL)."""
    blob = json.dumps(obj).encode()
    t = torch.frombuffer(bytearray(blob), dtype=torch.uint8).to(device)
    parts = _gather_variable_u8(t, device)
    return [json.loads(bytes(p.cpu().numpy()).decode()) for p in parts]


def merge_agg_partials(parts: List[Dict]) -> Dict:
    """Merge per-rank {key_rows, aggs, agg_ops} by agg op."""
    agg_ops = next((p["agg_ops"] for p in parts if p["aggs"]), [])
    merged: Dict[tuple, list] = {}
    keys_of: Dict[tuple, list] = {}
    for p in parts:
        for key, agg in zip(p["key_rows"], p["aggs"]):
            k = tuple(tuple(x) if isinstance(x, list) else x for x in key)
            acc = merged.get(k)
            if acc is None:
                merged[k] = list(agg)
                keys_of[k] = key
                continue
            for ai, op in enumerate(agg_ops):
                if op in (Q.AGGOP_COUNT, Q.AGGOP_SUM):
                    acc[ai] += agg[ai]
                elif op == Q.AGGOP_MIN:
                    acc[ai] = min(acc[ai], agg[ai])
                else:
                    acc[ai] = max(acc[ai], agg[ai])
    return {"key_rows": [keys_of[k] for k in merged],
            "aggs": list(merged.values())}


def merge_metric_rows(parts: List[List[Dict]], max_fields=("rrt_max",
                                                           "rtt_max")) -> List[Dict]:
    """Merge per-rank 1s rollup rows: sum additive fields, max the maxima —
    the time-bucket reduce across shards."""
    merged: Dict[tuple, Dict] = {}
    for rows in parts:
        for r in rows:
            key = tuple(sorted((k, v) for k, v in r.items()
                               if isinstance(v, str) or
                               k in ("time", "vtap_id", "l7_protocol",
                                     "response_status", "server_port",
                                     "l3_epc_id", "protocol")))
            acc = merged.get(key)
            if acc is None:
                merged[key] = dict(r)
                continue
            for k, v in r.items():
                if isinstance(v, str) or k in ("time", "vtap_id",
                                               "l7_protocol",
                                               "response_status",
                                               "server_port", "l3_epc_id",
                                               "protocol"):
                    continue
                if k in max_fields:
                    acc[k] = max(acc[k], v)
                else:
                    acc[k] = acc[k] + v
    return sorted(merged.values(),
                  key=lambda r: (r.get("time", 0), r.get("vtap_id", 0)))


class DistQueryEngine:
    """SPMD distributed querier: every rank calls query() with the same SQL
    and gets the merged global result."""

    def __init__(self, engine, device: str = "cpu"):
        self.engine = engine
        self.device = device

    def query(self, sql: str) -> Dict:
        partial = self.engine.query_partial(sql)
        parts = exchange_json(partial, self.device)
        if partial["kind"] == "rows":
            # row-table / select / show results: concatenate + dedup
            first = parts[0]["result"]
            cols = first["columns"]
            seen = []
            seen_set = set()
            for p in parts:
                for row in p["result"]["values"]:
                    kt = tuple(tuple(x) if isinstance(x, list) else x
                               for x in row)
                    if kt not in seen_set:
                        seen_set.add(kt)
                        seen.append(row)
            return {"columns": cols, "values": seen}
        merged = merge_agg_partials([p for p in parts])
        return self.engine.finalize_groups(sql, merged["key_rows"],
                                           merged["aggs"])
