"""Query engine: DF-SQL over the GPU-resident hot store.

The counterpart of the reference querier's CHEngine
(server/querier/engine/clickhouse/clickhouse.go) with ClickHouse replaced by
the in-HBM columnar store: parse -> Plan -> GPU group-by/select kernels ->
host-side hydration (SmartEncoding id -> name), ORDER BY / LIMIT.

Tables:
  l7_flow_log, l4_flow_log      -> GPU segment scans (flow_log DB)
  application / application.1s  -> 1s app rollup rows (flow_metrics DB)
  network / network.1s          -> 1s net rollup rows
"""
from __future__ import annotations

import ipaddress
import re
from typing import Dict, List, Optional

from ..store import l7_schema as S
from ..wire.const_enums import L7_PROTOCOL_NAMES
from . import spec as Q
from .executor import execute
from .sql import parse_sql, SqlError
from .tags import L7_TAGS, L7_METRICS, L4_TAGS, L4_METRICS, TagDef

STATUS_NAMES = {0: "Success", 1: "Not Exist", 2: "Error", 3: "Server Error",
                4: "Client Error"}

SRC_ROW = 100  # sentinel family: python row-table field

_FROM_RE = re.compile(r"\bfrom\s+`?([\w.]+)`?", re.IGNORECASE)


def rollup_rows(rows, bucket_s: int):
    """1s rollup rows -> coarser buckets (reference datasource 1m/1h
    AggregatingMergeTree MVs, ingester/datasource/handle.go:136-172):
    sums for additive fields, max for *_max fields."""
    out = {}
    for r in rows:
        t = (r["time"] // bucket_s) * bucket_s
        key_fields = tuple(sorted(
            (k, v) for k, v in r.items()
            if isinstance(v, str) or k in _ROLLUP_KEY_FIELDS))
        key = (t,) + key_fields
        acc = out.get(key)
        if acc is None:
            acc = dict(r)
            acc["time"] = t
            out[key] = acc
        else:
            for k, v in r.items():
                if k == "time" or isinstance(v, str):
                    continue
                if k.endswith("_max"):
                    acc[k] = max(acc[k], v)
                elif k not in _ROLLUP_KEY_FIELDS:
                    acc[k] = acc[k] + v
    return sorted(out.values(), key=lambda r: r["time"])


_ROLLUP_KEY_FIELDS = frozenset((
    "vtap_id", "l7_protocol", "response_status", "server_port", "l3_epc_id",
    "l3_epc_id_0", "l3_epc_id_1", "protocol", "tap_type", "acl_gid"))


def _row_tags(fields: List[str]) -> Dict[str, TagDef]:
    return {f: TagDef(f, SRC_ROW, i, hydrate="raw")
            for i, f in enumerate(fields)}


def _match_paren(sql: str, i: int) -> int:
    """`i` points at '('; return the index of the matching ')'.
    Skips over single-quoted string literals."""
    depth = 0
    in_str = False
    for j in range(i, len(sql)):
        c = sql[j]
        if in_str:
            if c == "'":
                in_str = False
        elif c == "'":
            in_str = True
        elif c == "(":
            depth += 1
        elif c == ")":
            depth -= 1
            if depth == 0:
                return j
    raise SqlError("unbalanced parentheses")


_STOP_KWS = {"where", "group", "order", "limit", "slimit", "having", "union"}


def split_with(sql: str):
    """'WITH a AS (...), b AS (...) SELECT ...' -> ([(name, inner)], main).
    Reference counterpart: CHEngine's WITH handling
    (server/querier/engine/clickhouse — TransWhere/with clauses)."""
    m = re.match(r"\s*with\s+", sql, re.IGNORECASE)
    pos = m.end()
    ctes = []
    while True:
        m = re.match(r"\s*`?(\w+)`?\s+as\s*", sql[pos:], re.IGNORECASE)
        if not m:
            raise SqlError("WITH: expected `name AS (...)`")
        name = m.group(1).lower()
        i = pos + m.end()
        if i >= len(sql) or sql[i] != "(":
            raise SqlError("WITH: expected '(' after AS")
        j = _match_paren(sql, i)
        ctes.append((name, sql[i + 1:j]))
        pos = j + 1
        m = re.match(r"\s*,", sql[pos:])
        if not m:
            break
        pos += m.end()
    return ctes, sql[pos:]


def _result_rows(res: Dict) -> List[Dict]:
    return [dict(zip(res["columns"], row)) for row in res["values"]]


def _plan_time_range(plan):
    """(lo_ns, hi_ns) from the plan's AND-level time predicates on
    start_time (SRC_U64 col 0), or None — drives cold-segment pruning."""
    lo, hi = None, None
    for t in plan.terms:
        if t.group != 0 or t.family != Q.SRC_U64 or t.idx != 0:
            continue
        if t.op == Q.OP_GE or t.op == Q.OP_GT:
            lo = t.v0 if lo is None else max(lo, t.v0)
        elif t.op == Q.OP_LE or t.op == Q.OP_LT:
            hi = t.v0 if hi is None else min(hi, t.v0)
        elif t.op == Q.OP_BETWEEN:
            lo = t.v0 if lo is None else max(lo, t.v0)
            hi = t.v1 if hi is None else min(hi, t.v1)
        elif t.op == Q.OP_EQ:
            lo = t.v0 if lo is None else max(lo, t.v0)
            hi = t.v0 if hi is None else min(hi, t.v0)
    if lo is None and hi is None:
        return None
    return (lo if lo is not None else 0,
            hi if hi is not None else (1 << 63))


def _plan_needed(plan) -> Optional[Dict]:
    """Column set a plan touches, by family — drives lazy cold-segment
    decompression. None = everything (row-fetch paths)."""
    if plan.select_rows:
        return None
    need: Dict[int, set] = {}

    def add(fam, idx):
        if fam == Q.SRC_TIME_BUCKET:
            need.setdefault(Q.SRC_U64, set()).add(0)
        elif fam in (Q.SRC_STR_HASH, Q.SRC_ATTR_MATCH, Q.SRC_ATTR_VAL):
            need.setdefault(fam, set())
        elif fam in (Q.SRC_U64, Q.SRC_U32, Q.SRC_U8, Q.SRC_DID, Q.SRC_KG):
            need.setdefault(fam, set()).add(idx)

    for t in plan.terms:
        add(t.family, t.idx)
    for k in plan.keys:
        add(k.family, k.idx)
    for a in plan.aggs:
        add(a.family, a.idx)
    for m in plan.agg_meta:
        if "family" in m:
            add(m["family"], m["idx"])
    return need


class QueryEngine:
    def __init__(self, pipeline, device: str = "cpu", l4_pipeline=None,
                 remote_hydrator=None):
        """pipeline: ingest.L7IngestPipeline; l4_pipeline optional."""
        self.pipe = pipeline
        self.l4 = l4_pipeline
        self.device = device
        self.remote = remote_hydrator
        # tagrecorder name maps (id -> display name), set by the server
        self.name_maps = {}
        # serializes query execution against concurrent ingest (the
        # server shares this lock with its ingest handlers) and protects
        # the shared cold-scratch segments from concurrent queries
        import threading
        self.lock = threading.RLock()

    # ----------------------------------------------------------- dispatch
    def query(self, sql: str, _ctes: Optional[Dict[str, Dict]] = None) -> Dict:
        with self.lock:
            try:
                # deferred ingest bookkeeping (async dictionary harvest)
                # must land before hydration reads the host maps
                for pipe in (self.pipe, self.l4):
                    if pipe is not None and hasattr(pipe, "sync_stats"):
                        pipe.sync_stats()
                return self._query_locked(sql, _ctes)
            finally:
                self._release_scratch()

    def _release_scratch(self) -> None:
        """Return cold-materialization scratch segments to the store's
        free-list once the query is finished (bounds query-time HBM)."""
        for pipe in (self.pipe, self.l4):
            if pipe is not None:
                pipe.segments.release_scratch()

    def _query_locked(self, sql: str,
                      _ctes: Optional[Dict[str, Dict]] = None) -> Dict:
        stripped = sql.strip().lower()
        if stripped.startswith("show"):
            return self._show(sql)
        if stripped.startswith("with"):
            ctes, main = split_with(sql)
            env = dict(_ctes or {})
            for name, inner in ctes:
                env[name] = self.query(inner, _ctes=env)
            return self.query(main, _ctes=env)
        md = re.search(r"\bfrom\s*\(", sql, re.IGNORECASE)
        if md:
            # derived table: FROM ( SELECT ... ) [AS alias]
            i = md.end() - 1
            j = _match_paren(sql, i)
            res = self.query(sql[i + 1:j], _ctes=_ctes)
            rest = sql[j + 1:]
            ma = re.match(r"\s*(?:as\s+)?`?(\w+)`?", rest, re.IGNORECASE)
            if ma and ma.group(1).lower() not in _STOP_KWS:
                rest = rest[ma.end():]
            outer = sql[: md.start()] + " FROM __sub__ " + rest
            return self._run_rows(outer, _result_rows(res), time_base_s=0)
        m = _FROM_RE.search(sql)
        table = m.group(1).lower() if m else "l7_flow_log"
        if _ctes and table in _ctes:
            return self._run_rows(sql, _result_rows(_ctes[table]),
                                  time_base_s=0)
        if table in ("l7_flow_log", "l7_flow_log.l7_flow_log"):
            plan = parse_sql(sql, dictionary=self.pipe.dict,
                             time_base_s=self.pipe.time_base_s,
                             tags=L7_TAGS, metrics=L7_METRICS,
                             name_maps=self.name_maps)
            return self._run_segments(
                plan, self.pipe.segments.scan_list(
                    needed=_plan_needed(plan),
                    time_range=_plan_time_range(plan)),
                L7_TAGS, S.STR_COLS)
        if table == "l4_flow_log":
            if self.l4 is None:
                raise SqlError("l4_flow_log table not enabled")
            plan = parse_sql(sql, dictionary=None,
                             time_base_s=self.l4.time_base_s,
                             tags=L4_TAGS, metrics=L4_METRICS,
                             name_maps=self.name_maps)
            from ..store import l4_schema as L4S
            return self._run_segments(
                plan, self.l4.segments.scan_list(
                    needed=_plan_needed(plan),
                    time_range=_plan_time_range(plan)),
                L4_TAGS, L4S.STR_COLS)
        row_tables = {
            "event": "event_rows", "perf_event": "perf_event_rows",
            "alert_event": "alert_event_rows",
            "application_log": "app_log_rows", "log": "app_log_rows",
            "trace_tree": "trace_tree_rows",
        }
        if table in row_tables:
            rows = getattr(self, row_tables[table], lambda: [])()
            return self._run_rows(sql, rows, time_base_s=0)
        if table == "application.agent":
            rows = getattr(self, "agent_app_rows", lambda: [])()
            return self._run_rows(sql, rows, time_base_s=self.pipe.time_base_s)
        if table == "network.agent":
            rows = getattr(self, "agent_net_rows", lambda: [])()
            return self._run_rows(sql, rows, time_base_s=self.pipe.time_base_s)
        if table.startswith("deepflow_system") or \
                table.startswith("deepflow_tenant"):
            rows = getattr(self, "system_rows", [])
            return self._run_rows(sql, rows, time_base_s=0)
        # flow_metrics table family: native GPU rollup tables
        # (network{,_map}.{1s,1m}, application{,_map}.{1s,1m},
        # traffic_policy.1m), coarser datasources derived from the 1m rows
        if table.split(".", 1)[0] in ("application", "application_map",
                                      "network", "network_map",
                                      "traffic_policy"):
            pipe = self.pipe if table.startswith("application") else self.l4
            if pipe is None:
                raise SqlError(f"{table} needs the l4 pipeline")
            base_name = table.split(".", 1)[0]
            native = pipe.rollups.get(table if "." in table
                                      else base_name + ".1s")
            if native is not None:
                rows = native.rows()
            else:
                iv = self._table_interval(table)
                src = pipe.rollups.get(base_name + (".1m" if iv % 60 == 0
                                                    else ".1s"))
                if src is None:
                    raise SqlError(f"unknown table {table!r}")
                rows = rollup_rows(src.rows(), iv)
            return self._run_rows(sql, rows, time_base_s=pipe.time_base_s)
        raise SqlError(f"unknown table {table!r}")

    # datasource intervals (reference ingester/datasource REST: 1h/1d MVs)
    DATASOURCE_INTERVALS = {"1s": 1, "1m": 60, "1h": 3600, "1d": 86400}

    def _table_interval(self, table: str) -> int:
        if "." not in table:
            return 1
        suffix = table.split(".", 1)[1]
        extra = getattr(self, "extra_datasources", {})
        if suffix in extra:
            return extra[suffix]
        iv = self.DATASOURCE_INTERVALS.get(suffix)
        if iv is None:
            raise SqlError(f"unknown datasource interval {suffix!r}")
        return iv

    def add_datasource(self, name: str, interval_s: int) -> None:
        if not hasattr(self, "extra_datasources"):
            self.extra_datasources = {}
        self.extra_datasources[name] = interval_s

    # ----------------------------------------------------------- show
    def _show(self, sql: str) -> Dict:
        parts = sql.strip().split()
        what = parts[1].lower() if len(parts) > 1 else ""
        m = _FROM_RE.search(sql)
        table = m.group(1).lower() if m else "l7_flow_log"
        tags = L4_TAGS if table == "l4_flow_log" else L7_TAGS
        mets = L4_METRICS if table == "l4_flow_log" else L7_METRICS
        # "show tag <name> values [from <table>]" — flow_tag discovery
        # (reference flow_tag custom_field_value tables)
        if what == "tag" and len(parts) >= 4 and parts[3].lower() == "values":
            name = parts[2].strip("`")
            td = tags.get(name)
            vals = []
            if td is not None and td.hydrate.startswith("dict:"):
                dom = int(td.hydrate.split(":")[1])
                vals = sorted(s.decode("utf-8", "replace")
                              for (d, s) in self.pipe.dict.str_to_id
                              if d == dom)
            elif name == "attribute_names":
                vals = sorted(s.decode("utf-8", "replace")
                              for (d, s) in self.pipe.dict.str_to_id
                              if d == 6)
            elif name.startswith("attribute."):
                vals = sorted(s.decode("utf-8", "replace")
                              for (d, s) in self.pipe.dict.str_to_id
                              if d == 7)
            return {"columns": ["value"], "values": [[v] for v in vals]}
        if what in ("tags", "metrics"):
            # db_descriptions-backed discovery (reference:
            # querier/db_descriptions/clickhouse/{tag,metrics}/)
            from .descriptions import table_descriptions
            catalog = table_descriptions(self)
            entry = catalog.get(table)
            if entry is None and "." not in table:
                entry = catalog.get(table + ".1s")
            cols = ["name", "display_name", "unit", "type", "description"]
            if entry is not None:
                rows_ = entry[what]
                return {"columns": cols,
                        "values": [[d["name"], d["display_name"],
                                    d["unit"], d["type"], d["description"]]
                                   for d in rows_]}
            # unknown table: fall back to the engine tag map
            from .descriptions import describe
            src = tags if what == "tags" else mets
            return {"columns": cols,
                    "values": [[d["name"], d["display_name"], d["unit"],
                                d["type"], d["description"]]
                               for d in (describe(n, "tag" if what == "tags"
                                                  else "metric")
                                         for n in sorted(src))]}
        if what == "tables":
            names = ["l7_flow_log", "l4_flow_log"]
            for pipe in (self.pipe, self.l4):
                if pipe is not None and hasattr(pipe, "rollups"):
                    names.extend(sorted(pipe.rollups.tables))
            return {"columns": ["name"], "values": [[n] for n in names]}
        raise SqlError(f"unsupported show: {sql!r}")

    # --------------------------------------------------- distributed hook
    def query_partial(self, sql: str) -> Dict:
        """Shard-local partial for distributed merge: aggregate queries
        return hydrated group keys + RAW agg vectors (avg still split as
        sum+count so cross-shard merge is exact); everything else returns
        the finished local result under kind='rows'."""
        with self.lock:
            try:
                return self._query_partial_locked(sql)
            finally:
                self._release_scratch()

    def _query_partial_locked(self, sql: str) -> Dict:
        stripped = sql.strip().lower()
        m = _FROM_RE.search(sql)
        table = m.group(1).lower() if m else "l7_flow_log"
        if stripped.startswith(("show", "with")) or \
                re.search(r"\bfrom\s*\(", sql, re.IGNORECASE) or \
                table not in ("l7_flow_log", "l4_flow_log"):
            # WITH/derived queries run shard-locally and merge as rows:
            # correct when the inner query is per-row (filters/select);
            # a cross-shard inner aggregate needs the dist engine's agg
            # path (round-2: rewrite WITH into partial+finalize).
            return {"kind": "rows", "result": self.query(sql)}
        if table == "l4_flow_log" and self.l4 is None:
            raise SqlError("l4_flow_log table not enabled")
        if table == "l7_flow_log":
            plan = parse_sql(sql, dictionary=self.pipe.dict,
                             time_base_s=self.pipe.time_base_s,
                             tags=L7_TAGS, metrics=L7_METRICS,
                             name_maps=self.name_maps)
            segments, tags, str_cols = (self.pipe.segments.scan_list(
                needed=_plan_needed(plan),
                time_range=_plan_time_range(plan)), L7_TAGS,
                                        S.STR_COLS)
        else:
            from ..store import l4_schema as L4S
            plan = parse_sql(sql, dictionary=None,
                             time_base_s=self.l4.time_base_s,
                             tags=L4_TAGS, metrics=L4_METRICS,
                             name_maps=self.name_maps)
            segments, tags, str_cols = (self.l4.segments.scan_list(
                needed=_plan_needed(plan),
                time_range=_plan_time_range(plan)), L4_TAGS,
                                        L4S.STR_COLS)
        if plan.select_rows:
            return {"kind": "rows",
                    "result": self._run_select(plan, segments, tags,
                                               str_cols)}
        groups = execute(plan, segments, self.device, kg=self.pipe.kg)
        key_rows = []
        aggs = []
        raw_keys = []
        for g in groups:
            key_rows.append([self._hydrate(meta["hydrate"], g["key"][ki])
                             for ki, meta in enumerate(plan.key_meta)])
            aggs.append(g["agg"][: len(plan.aggs)])
            raw_keys.append(g["key"])
        out = {
            "kind": "agg",
            "key_rows": key_rows,
            "aggs": aggs,
            "agg_ops": [a.op for a in plan.aggs],
        }
        q_metas = [m for m in plan.agg_meta
                   if m["op"] in ("percentile", "apdex")]
        if q_metas:
            # cross-shard quantiles merge as log-bucket histograms
            # (16 sub-buckets per octave: <=~4.5% relative value error —
            # the quantileTiming-style tradeoff); apdex stays exact by
            # shipping (satisfied, tolerated, total) counts per group.
            out["qhist"] = self._quantile_partial(plan, segments, q_metas,
                                                  raw_keys)
        return out

    _QH_SCALE = 16  # histogram sub-buckets per value octave

    def _qbucket_value(self, b: int) -> float:
        return 2.0 ** (b / self._QH_SCALE) - 1.0

    def _quantile_partial(self, plan, segments, q_metas, raw_keys):
        """Per-group histograms / apdex counts aligned with key_rows."""
        import math
        from .executor import execute_grouped_values
        uniq, per_meta = execute_grouped_values(plan, segments, q_metas,
                                                self.device,
                                                kg=self.pipe.kg)
        index = {tuple(int(x) & ((1 << 64) - 1) for x in uniq[i].tolist()):
                 i for i in range(uniq.shape[0])}
        out = []
        for key in raw_keys:
            gi = index.get(tuple(int(x) & ((1 << 64) - 1) for x in key))
            row = []
            for mi, meta in enumerate(q_metas):
                if gi is None or mi not in per_meta:
                    row.append(None)
                    continue
                svals, starts, counts = per_meta[mi]
                s0, c0 = int(starts[gi]), int(counts[gi])
                vals = svals[s0:s0 + c0]
                if meta["op"] == "apdex":
                    t = float(meta.get("param", 100000))
                    sat = int((vals <= t).sum())
                    tol = int(((vals > t) & (vals <= 4 * t)).sum())
                    row.append({"apdex": [sat, tol, int(vals.numel())]})
                else:
                    hist = {}
                    for v in vals.tolist():
                        b = int(round(math.log2(v + 1.0) * self._QH_SCALE))
                        hist[b] = hist.get(b, 0) + 1
                    row.append({"hist": hist})
            out.append(row)
        return out

    def finalize_groups(self, sql: str, key_rows, aggs,
                        qdata=None) -> Dict:
        """Turn merged (hydrated keys, raw aggs) back into a result table
        using the local plan (column names, avg division, order/limit).
        `qdata` carries merged quantile histograms / apdex counts per
        group (aligned with key_rows), from the distributed merge."""
        m = _FROM_RE.search(sql)
        table = m.group(1).lower() if m else "l7_flow_log"
        if table == "l4_flow_log":
            plan = parse_sql(sql, dictionary=None,
                             time_base_s=self.l4.time_base_s,
                             tags=L4_TAGS, metrics=L4_METRICS,
                             name_maps=self.name_maps)
        else:
            plan = parse_sql(sql, dictionary=self.pipe.dict,
                             time_base_s=self.pipe.time_base_s,
                             tags=L7_TAGS, metrics=L7_METRICS,
                             name_maps=self.name_maps)
        columns = plan.key_names + plan.agg_names
        rows = []
        for gi, (key, agg) in enumerate(zip(key_rows, aggs)):
            row = list(key)
            ai = 0
            qi = 0
            for meta in plan.agg_meta:
                if meta["op"] == "avg":
                    ssum, cnt = agg[ai], agg[ai + 1]
                    ai += 2
                    row.append(ssum / cnt if cnt else None)
                elif meta["op"] in ("percentile", "apdex"):
                    q = qdata[gi][qi] if qdata and qdata[gi] else None
                    qi += 1
                    ai += 1
                    row.append(self._finish_qdata(meta, q))
                else:
                    row.append(agg[ai])
                    ai += 1
            rows.append(row)
        if plan.slimit:
            rows = self._apply_slimit(plan, rows)
        rows = self._order_limit(plan, columns, rows)
        return {"columns": columns, "values": rows}

    def _finish_qdata(self, meta, q):
        """Merged quantile data -> final value (histogram walk with
        within-bucket interpolation, or exact apdex counts)."""
        if q is None:
            return None
        if "apdex" in q:
            sat, tol, total = q["apdex"]
            return (sat + tol / 2) / total if total else None
        hist = {int(k): v for k, v in q["hist"].items()}
        total = sum(hist.values())
        if total == 0:
            return None
        qq = float(meta.get("param", 95)) / 100.0
        target = qq * (total - 1)
        seen = 0.0
        for b in sorted(hist):
            c = hist[b]
            if seen + c > target:
                # interpolate inside the bucket's value span
                lo = self._qbucket_value(b - 1) if b > 0 else 0.0
                hi = self._qbucket_value(b)
                frac = (target - seen) / c
                return lo + (hi - lo) * frac if c > 1 else hi
            seen += c
        return self._qbucket_value(max(hist))

    # ----------------------------------------------------------- segments
    def _run_segments(self, plan: Q.Plan, segments, tags, str_cols,
                      kg=None) -> Dict:
        kg = kg if kg is not None else self.pipe.kg
        if plan.select_rows:
            return self._run_select(plan, segments, tags, str_cols, kg=kg)
        import os as _os
        import time as _time
        _qt = _os.environ.get("DF_QTIME")
        _t0 = _time.perf_counter()
        groups = execute(plan, segments, self.device, kg=kg)
        if _qt:
            import sys as _sys
            print(f"[qtime] execute {1e3*(_time.perf_counter()-_t0):.2f}ms"
                  f" groups={len(groups)}", file=_sys.stderr)
        q_metas = [m for m in plan.agg_meta
                   if m["op"] in ("percentile", "apdex")]
        q_lookup = None
        if q_metas:
            # single-pass grouped gather (no per-group re-scan, no group
            # cap): values sorted within each group for quantile math
            from .executor import execute_grouped_values
            uniq, per_meta = execute_grouped_values(plan, segments,
                                                    q_metas, self.device,
                                                    kg=kg)
            key_index = {tuple(int(x) & ((1 << 64) - 1)
                               for x in uniq[i].tolist()): i
                         for i in range(uniq.shape[0])}
            q_lookup = (key_index, per_meta,
                        {id(m): mi for mi, m in enumerate(q_metas)})
        columns = plan.key_names + plan.agg_names
        rows: List[List] = []
        if _qt:
            _t0 = _time.perf_counter()
        # hoist hydration dispatch out of the per-group loop: closures
        # bound to the host maps, no per-cell string matching
        hyds = [self._hydrator(meta["hydrate"]) for meta in plan.key_meta]
        for g in groups:
            gk = g["key"]
            row = [hyds[ki](gk[ki]) for ki in range(len(hyds))]
            ai = 0
            for meta in plan.agg_meta:
                if meta["op"] == "avg":
                    ssum = g["agg"][ai]
                    cnt = g["agg"][ai + 1]
                    ai += 2
                    row.append(ssum / cnt if cnt else None)
                elif meta["op"] in ("percentile", "apdex"):
                    key_index, per_meta, meta_ix = q_lookup
                    gi = key_index.get(tuple(int(x) & ((1 << 64) - 1)
                                             for x in g["key"]))
                    ai += 1
                    if gi is None:
                        row.append(None)
                    else:
                        mi = meta_ix[id(meta)]
                        svals, starts, counts = per_meta[mi]
                        s0 = int(starts[gi])
                        c0 = int(counts[gi])
                        row.append(self._finish_quantile(
                            meta, svals[s0:s0 + c0]))
                else:
                    row.append(g["agg"][ai])
                    ai += 1
            rows.append(row)
        if _qt:
            import sys as _sys
            print(f"[qtime] assemble {1e3*(_time.perf_counter()-_t0):.2f}"
                  f"ms rows={len(rows)}", file=_sys.stderr)
            _t0 = _time.perf_counter()
        if plan.slimit:
            rows = self._apply_slimit(plan, rows)
        rows = self._order_limit(plan, columns, rows)
        if _qt:
            import sys as _sys
            print(f"[qtime] order {1e3*(_time.perf_counter()-_t0):.2f}ms",
                  file=_sys.stderr)
        return {"columns": columns, "values": rows}

    @staticmethod
    def _finish_quantile(meta, vals):
        import torch
        if vals is None or vals.numel() == 0:
            return None
        if meta["op"] == "percentile":
            q = float(meta.get("param", 95)) / 100.0
            return float(torch.quantile(vals, q))
        # Apdex: param = satisfied threshold (us); tolerated = 4x
        t = float(meta.get("param", 100000))
        sat = float((vals <= t).sum())
        tol = float(((vals > t) & (vals <= 4 * t)).sum())
        return (sat + tol / 2) / vals.numel()

    def _apply_slimit(self, plan: Q.Plan, rows: List[List]) -> List[List]:
        """Keep only the top-N series (distinct non-time key combos), ranked
        by the first aggregate (reference SLIMIT semantics)."""
        non_time = [i for i, n in enumerate(plan.key_names) if n != "time"]
        first_agg = len(plan.key_names)
        series: Dict[tuple, float] = {}
        for r in rows:
            k = tuple(r[i] for i in non_time)
            v = r[first_agg] if len(r) > first_agg and \
                isinstance(r[first_agg], (int, float)) else 0
            series[k] = series.get(k, 0) + (v or 0)
        top = sorted(series, key=lambda k: -series[k])[: plan.slimit]
        keep = set(top)
        return [r for r in rows if tuple(r[i] for i in non_time) in keep]

    # ----------------------------------------------------------- select
    def _run_select(self, plan: Q.Plan, segments, tags, str_cols,
                    kg=None) -> Dict:
        hits = execute(plan, segments, self.device,
                       kg=kg if kg is not None else self.pipe.kg)
        cols = plan.select_cols
        if cols == ["*"]:
            cols = ["start_time", "end_time", "flow_id", "l7_protocol",
                    "request_domain", "request_resource", "response_status",
                    "response_code", "response_duration", "trace_id",
                    "span_id", "service_name"] \
                if "trace_id" in str_cols else \
                ["start_time", "end_time", "flow_id", "protocol",
                 "byte_tx", "byte_rx", "rtt", "close_type"]
        rows = []
        for si, r in hits:
            seg = segments[si]
            rows.append([self._fetch(seg, r, c, tags, str_cols) for c in cols])
        if plan.limit:
            rows = rows[: plan.limit]
        return {"columns": cols, "values": rows}

    def _fetch(self, seg, row: int, col: str, tags, str_cols):
        if col.startswith("attribute."):
            # custom tag by name: scan the row's attr-id slots
            # (layout: pool[start .. start+cnt) = name ids,
            #  pool[start+cnt .. start+2cnt) = value ids)
            from ..store.l7_schema import (DICT_DOM_ATTR_NAME,
                                           DICT_DOM_ATTR_VALUE)
            want = self.pipe.dict.lookup_id(DICT_DOM_ATTR_NAME,
                                            col[len("attribute."):].encode())
            if want is None:
                return None
            cnt = int(seg.attr_cnt[row])
            start = int(seg.attr_start[row])
            for i in range(cnt):
                nid = int(seg.attr_pool[start + i]) & 0xFFFFFFFF
                if nid == want:
                    vid = int(seg.attr_pool[start + cnt + i]) & 0xFFFFFFFF
                    return self.pipe.dict.hydrate(DICT_DOM_ATTR_VALUE,
                                                  [vid])[0]
            return None
        if col in ("trace_id", "span_id") and hasattr(seg, "attr_pool"):
            # binary-transcoded ids: reconstruct the hex form (pool holds
            # only the non-hex fallbacks)
            hi_i = S.U64_COLS.index("trace_id_hi")
            lo_i = S.U64_COLS.index("trace_id_lo")
            sp_i = S.U64_COLS.index("span_id_b")
            M = (1 << 64) - 1
            if col == "trace_id":
                hi, lo = int(seg.u64[hi_i, row]) & M, \
                    int(seg.u64[lo_i, row]) & M
                if hi | lo:
                    return f"{hi:016x}{lo:016x}"
            else:
                sv = int(seg.u64[sp_i, row]) & M
                if sv:
                    return f"{sv:016x}"
            # fall through to the pooled fallback string
        td = tags.get(col)
        if td is not None and td.family != Q.SRC_TRACE128:
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
                # query-time KG join (ids not materialized per row)
                from ..store import l7_schema as S7
                side, j = idx // S7.N_KG, idx % S7.N_KG
                epc = int(seg.u32[3 + side, row]) & 0xFFFFFFFF
                ip = int(seg.u32[1 + side, row]) & 0xFFFFFFFF
                info = self.pipe.kg.host.get((epc, ip))
                return self._hydrate(td.hydrate,
                                     info.as_list()[j] if info else 0)
        if col in str_cols:
            # pooled string columns: row block ref + per-col u16 lens
            # (dict-encoded string tags were already handled via SRC_DID)
            if hasattr(seg, "attr_pool"):  # l7 segment: map via POOL_POS
                if col not in S.POOL_POS:
                    raise SqlError(f"column {col!r} is dict-encoded; "
                                   f"select it via its tag")
                sidx = S.POOL_POS[col]
            else:
                sidx = str_cols.index(col)
            rr = int(seg.str_rowref[row]) & ((1 << 64) - 1)
            ln = int(seg.str_lens[sidx, row]) & 0xFFFF
            if ln == 0:
                return ""
            off = (rr >> 16) + sum(
                int(seg.str_lens[c, row]) & 0xFFFF for c in range(sidx))
            raw = bytes(seg.pool[off:off + ln].numpy()
                        if seg.pool.device.type == "cpu"
                        else seg.pool[off:off + ln].cpu().numpy())
            if col in ("ip6_0", "ip6_1") and ln == 16:
                import ipaddress
                return str(ipaddress.IPv6Address(raw))
            return raw.decode("utf-8", "replace")
        raise SqlError(f"unknown select column {col!r}")

    # ----------------------------------------------------------- row tables
    def _run_rows(self, sql: str, rows: List[Dict],
                  time_base_s: int) -> Dict:
        if not rows:
            return {"columns": [], "values": []}
        fields = list(rows[0].keys())
        tags = _row_tags(fields)
        mets = {f: tags[f] for f in fields}
        mets["log_count"] = TagDef("log_count", Q.SRC_CONST0, 0)
        plan = parse_sql(sql, dictionary=None, time_base_s=time_base_s,
                         tags=tags, metrics=mets)
        if plan.impossible:
            return {"columns": [], "values": []}

        def val(r: Dict, fam: int, idx: int, bucket: int = 0):
            if fam == SRC_ROW:
                return r[fields[idx]]
            if fam == Q.SRC_TIME_BUCKET:
                rel = r.get("time", 0) - time_base_s
                if bucket:
                    rel = (rel // bucket) * bucket
                return time_base_s + rel
            if fam == Q.SRC_U64 and idx == 0:  # `time >= n` filter shape
                return r.get("time", 0) * 10**9
            return 0

        OPS = {Q.OP_EQ: lambda a, b: a == b, Q.OP_NE: lambda a, b: a != b,
               Q.OP_LT: lambda a, b: a < b, Q.OP_LE: lambda a, b: a <= b,
               Q.OP_GT: lambda a, b: a > b, Q.OP_GE: lambda a, b: a >= b,
               Q.OP_BETWEEN: lambda a, b: True}

        def term_ok(r, t):
            if t.family == Q.SRC_CONST0:
                return OPS[t.op](0, t.v0)
            v = val(r, t.family, t.idx)
            if t.op == Q.OP_BETWEEN:
                return t.v0 <= v <= t.v1
            try:
                return OPS[t.op](v, t.v0)
            except TypeError:
                return False
        # CNF: group 0 terms are ANDed; each group>=1 is an OR clause
        filtered = []
        for r in rows:
            ok = all(term_ok(r, t) for t in plan.terms if t.group == 0)
            if ok:
                groups_seen: Dict[int, bool] = {}
                for t in plan.terms:
                    if t.group:
                        groups_seen[t.group] = groups_seen.get(
                            t.group, False) or term_ok(r, t)
                ok = all(groups_seen.values())
            if ok:
                filtered.append(r)
        if plan.select_rows:
            cols = plan.select_cols
            if cols == ["*"]:
                cols = fields
            out = [[r.get(c) for c in cols] for r in filtered]
            if plan.limit:
                out = out[: plan.limit]
            return {"columns": cols, "values": out}
        groups: Dict[tuple, list] = {}
        for r in filtered:
            key = tuple(val(r, k.family, k.idx, k.bucket) for k in plan.keys)
            acc = groups.setdefault(key, [None] * len(plan.aggs))
            for ai, a in enumerate(plan.aggs):
                v = 1 if a.op == Q.AGGOP_COUNT else val(r, a.family, a.idx)
                cur = acc[ai]
                if a.op in (Q.AGGOP_COUNT, Q.AGGOP_SUM):
                    acc[ai] = (cur or 0) + v
                elif a.op == Q.AGGOP_MIN:
                    acc[ai] = v if cur is None else min(cur, v)
                else:
                    acc[ai] = v if cur is None else max(cur, v)
        columns = plan.key_names + plan.agg_names
        out = []
        for key, acc in groups.items():
            row = list(key)
            ai = 0
            for meta in plan.agg_meta:
                if meta["op"] == "avg":
                    s_, c_ = acc[ai] or 0, acc[ai + 1] or 0
                    ai += 2
                    row.append(s_ / c_ if c_ else None)
                else:
                    row.append(acc[ai])
                    ai += 1
            out.append(row)
        out = self._order_limit(plan, columns, out)
        return {"columns": columns, "values": out}

    # ----------------------------------------------------------- hydrate
    def _hydrator(self, how: str):
        """Column-level hydration closure (bound maps, no per-cell
        dispatch) — the per-cell _hydrate path costs ~3us x cells on
        many-group results."""
        if how == "int":
            return lambda v: v
        if how == "time":
            tb = self.pipe.time_base_s
            return lambda v: tb + v
        if how.startswith("dict:"):
            dom = int(how.split(":")[1])
            m = self.pipe.dict.id_to_str
            from ..store import l7_schema as _S
            inv = _S.DICT_ID_INVALID

            def f(v, m=m, dom=dom, inv=inv):
                if v < 0 or v == inv:
                    return None
                b = m.get((dom, v & 0xFFFFFFFF))
                return b.decode("utf-8", "replace") if b is not None \
                    else None
            return f
        if how == "l7proto":
            return lambda v: L7_PROTOCOL_NAMES.get(v, str(v))
        if how == "status":
            return lambda v: STATUS_NAMES.get(v, str(v))
        return lambda v, how=how: self._hydrate(how, v)

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
        if how.startswith("kgname:"):
            mp = how.split(":", 1)[1]
            return self.name_maps.get(mp, {}).get(v, str(v))
        return v

    _HAVING_OPS = {"=": lambda a, b: a == b, "!=": lambda a, b: a != b,
                   "<>": lambda a, b: a != b,
                   "<": lambda a, b: a < b, "<=": lambda a, b: a <= b,
                   ">": lambda a, b: a > b, ">=": lambda a, b: a >= b}

    def _order_limit(self, plan: Q.Plan, columns, rows):
        if plan.having:
            for name, op, num in plan.having:
                if name not in columns:
                    from .sql import SqlError
                    raise SqlError(f"HAVING references unknown column "
                                   f"{name!r}")
                ci = columns.index(name)
                fn = self._HAVING_OPS[op]
                rows = [r for r in rows
                        if r[ci] is not None and fn(r[ci], num)]
        if plan.order_by:
            for name, desc in reversed(plan.order_by):
                if name in columns:
                    i = columns.index(name)
                    rows.sort(key=lambda r: (r[i] is None, r[i]),
                              reverse=desc)
        else:
            rows.sort(key=lambda r: tuple(
                (x is None, str(type(x)), x) for x in
                r[: len(plan.key_names)]))
        if plan.limit:
            rows = rows[: plan.limit]
        return rows
