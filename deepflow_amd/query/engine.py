"""Query engine: DF-SQL over the GPU-resident hot store.

The counterpart of the reference querier's CHEngine
(server/querier/engine/clickhouse/clickhouse.go) with ClickHouse replaced by
the in-HBM columnar store: parse -> Plan -> GPU group-by/select kernels ->
host-side hydration (SmartEncoding id -> name), ORDER BY / LIMIT.
"""
from __future__ import annotations

import ipaddress
from typing import Dict, List, Optional

from ..store import l7_schema as S
from ..wire.const_enums import L7_PROTOCOL_NAMES
from . import spec as Q
from .executor import execute
from .sql import parse_sql, SqlError
from .tags import L7_TAGS, L7_METRICS

STATUS_NAMES = {0: "Success", 1: "Not Exist", 2: "Error", 3: "Server Error",
                4: "Client Error"}


class QueryEngine:
    def __init__(self, pipeline, device: str = "cpu",
                 remote_hydrator=None):
        """pipeline: ingest.L7IngestPipeline (owns segments/dict/kg/metrics).
        remote_hydrator: optional parallel.DictSync for cross-shard names."""
        self.pipe = pipeline
        self.device = device
        self.remote = remote_hydrator

    # ----------------------------------------------------------- dispatch
    def query(self, sql: str) -> Dict:
        stripped = sql.strip().lower()
        if stripped.startswith("show"):
            return self._show(sql)
        plan = parse_sql(sql, dictionary=self.pipe.dict,
                         time_base_s=self.pipe.time_base_s)
        if plan.select_rows:
            return self._run_select(plan)
        return self._run_agg(plan)

    # ----------------------------------------------------------- show
    def _show(self, sql: str) -> Dict:
        parts = sql.strip().split()
        what = parts[1].lower() if len(parts) > 1 else ""
        if what == "tags":
            cols = ["name", "display_name", "type"]
            vals = [[n, n, t.hydrate] for n, t in sorted(L7_TAGS.items())]
            return {"columns": cols, "values": vals}
        if what == "metrics":
            cols = ["name", "display_name", "type"]
            vals = [[n, n, "counter"] for n in sorted(L7_METRICS)]
            return {"columns": cols, "values": vals}
        if what == "tables":
            return {"columns": ["name"],
                    "values": [["l7_flow_log"], ["application.1s"]]}
        raise SqlError(f"unsupported show: {sql!r}")

    # ----------------------------------------------------------- agg
    def _run_agg(self, plan: Q.Plan) -> Dict:
        segments = self.pipe.segments.segments
        groups = execute(plan, segments, self.device)
        columns = plan.key_names + plan.agg_names
        rows: List[List] = []
        for g in groups:
            row = []
            for ki, meta in enumerate(plan.key_meta):
                row.append(self._hydrate(meta["hydrate"], g["key"][ki]))
            ai = 0
            for meta in plan.agg_meta:
                if meta["op"] == "avg":
                    ssum = g["agg"][ai]
                    cnt = g["agg"][ai + 1]
                    ai += 2
                    row.append(ssum / cnt if cnt else None)
                else:
                    row.append(g["agg"][ai])
                    ai += 1
            rows.append(row)
        rows = self._order_limit(plan, columns, rows)
        return {"columns": columns, "values": rows}

    # ----------------------------------------------------------- select
    def _run_select(self, plan: Q.Plan) -> Dict:
        segments = self.pipe.segments.segments
        hits = execute(plan, segments, self.device)
        cols = plan.select_cols
        if cols == ["*"]:
            cols = ["start_time", "end_time", "flow_id", "l7_protocol",
                    "request_domain", "request_resource", "response_status",
                    "response_code", "response_duration", "trace_id",
                    "span_id", "service_name"]
        rows = []
        for si, r in hits:
            seg = segments[si]
            rows.append([self._fetch(seg, r, c) for c in cols])
        if plan.limit:
            rows = rows[: plan.limit]
        return {"columns": cols, "values": rows}

    def _fetch(self, seg, row: int, col: str):
        if col in L7_TAGS:
            td = L7_TAGS[col]
            fam, idx = td.family, td.idx
            if fam == Q.SRC_U64:
                return int(seg.u64[idx, row])
            if fam == Q.SRC_U32:
                v = int(seg.u32[idx, row]) & 0xFFFFFFFF
                return self._hydrate(td.hydrate, v)
            if fam == Q.SRC_U8:
                return self._hydrate(td.hydrate, int(seg.u8[idx, row]))
            if fam == Q.SRC_DID:
                return self._hydrate(td.hydrate,
                                     int(seg.did[idx, row]) & 0xFFFFFFFF)
            if fam == Q.SRC_KG:
                return int(seg.kg[idx, row])
        if col in S.STR_COLS:
            sidx = S.STR_COLS.index(col)
            r = int(seg.strref[sidx, row]) & ((1 << 64) - 1)
            off, ln = r >> 16, r & 0xFFFF
            if ln == 0:
                return ""
            if seg.pool.device.type == "cpu":
                return bytes(seg.pool[off:off + ln].numpy()).decode(
                    "utf-8", "replace")
            return bytes(seg.pool[off:off + ln].cpu().numpy()).decode(
                "utf-8", "replace")
        raise SqlError(f"unknown select column {col!r}")

    # ----------------------------------------------------------- hydrate
    def _hydrate(self, how: str, v: int):
        if how == "int":
            return v
        if how == "time":
            return self.pipe.time_base_s + v
        if how.startswith("dict:"):
            dom = int(how.split(":")[1])
            s = self.pipe.dict.hydrate(dom, [v])[0]
            return s
        if how == "ip":
            return str(ipaddress.IPv4Address(v & 0xFFFFFFFF))
        if how == "l7proto":
            return L7_PROTOCOL_NAMES.get(v, str(v))
        if how == "status":
            return STATUS_NAMES.get(v, str(v))
        return v

    def _order_limit(self, plan: Q.Plan, columns, rows):
        if plan.order_by:
            for name, desc in reversed(plan.order_by):
                if name in columns:
                    i = columns.index(name)
                    rows.sort(key=lambda r: (r[i] is None, r[i]),
                              reverse=desc)
        else:
            rows.sort(key=lambda r: tuple(
                (x is None, x) for x in r[: len(plan.key_names)]))
        if plan.limit:
            rows = rows[: plan.limit]
        return rows
