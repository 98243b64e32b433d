"""PromQL engine over the rollup tables (reference: server/querier/app/
prometheus — PromQL -> DF-SQL conversion, converters.go).

Supported subset:
  selector:      metric{label="v",...}
  range fns:     rate(sel[w]) / increase(sel[w]) / avg_over_time(sel[w])
  aggregation:   sum|avg|min|max|count [by (l1, l2)] (expr)
  APIs:          /prom/api/v1/query (instant), /query_range, /series,
                 /label/<name>/values

Metric namespace (prom name -> table field):
  application_<field>  -> application.1s rollup (request, response,
                          client_error, server_error, rrt_sum, rrt_count,
                          rrt_max)
  network_<field>      -> network.1s rollup (byte_tx, byte_rx, packet_tx,
                          packet_rx, new_flow, closed_flow, rtt_sum, ...)
Buckets are per-second deltas, so rate(x[w]) = sum(window)/w and
increase(x[w]) = sum(window).
"""
from __future__ import annotations

import re
import time as _time
from typing import Dict, List, Optional, Tuple

from fastapi import Request

_SEL_RE = re.compile(
    r"^\s*(?P<name>[a-zA-Z_:][a-zA-Z0-9_:]*)\s*(?:\{(?P<matchers>[^}]*)\})?"
    r"\s*(?:\[(?P<range>\d+)(?P<runit>[smh])\])?\s*$")
def _balanced(s: str):
    """s starts with '('; return (inner, tail-after-close)."""
    depth = 0
    for i, ch in enumerate(s):
        if ch == "(":
            depth += 1
        elif ch == ")":
            depth -= 1
            if depth == 0:
                return s[1:i], s[i + 1:]
    return None, None


def _match_agg(expr: str):
    m = re.match(r"^\s*(sum|avg|min|max|count)\s*", expr)
    if not m:
        return None
    func, rest = m.group(1), expr[m.end():]
    by = None
    m2 = re.match(r"^by\s*\(([^)]*)\)\s*", rest)
    if m2:
        by = m2.group(1)
        rest = rest[m2.end():]
    if not rest.startswith("("):
        return None
    inner, tail = _balanced(rest)
    if inner is None:
        return None
    m3 = re.match(r"^\s*by\s*\(([^)]*)\)\s*$", tail)
    if m3:
        by = m3.group(1)
    elif tail.strip():
        return None
    return func, by, inner
_FN_RE = re.compile(
    r"^\s*(?P<fn>rate|irate|increase|avg_over_time|sum_over_time|"
    r"max_over_time)\s*\((?P<inner>.*)\)\s*$", re.DOTALL)

_UNIT = {"s": 1, "m": 60, "h": 3600}


class PromError(ValueError):
    pass


def _apply_op(a: float, b: float, op: str) -> float:
    if op == "+":
        return a + b
    if op == "-":
        return a - b
    if op == "*":
        return a * b
    return a / b if b else float("nan")


def _parse_matchers(s: Optional[str]) -> List[Tuple[str, str, str]]:
    out = []
    if not s:
        return out
    for part in re.split(r",\s*", s.strip()):
        if not part:
            continue
        m = re.match(r'([a-zA-Z_][a-zA-Z0-9_]*)\s*(=~|!=|=)\s*"([^"]*)"', part)
        if not m:
            raise PromError(f"bad matcher {part!r}")
        out.append((m.group(1), m.group(2), m.group(3)))
    return out


class PromQLEngine:
    DEFAULT_LOOKBACK = 300

    # flow_log_* metrics execute on the GPU store through the DF-SQL
    # engine (reference: PromQL -> SQL conversion, app/prometheus/service/
    # converters.go:1701; BASELINE config #4 — the group-by runs
    # k_query_agg, not a host row scan)
    FLOW_LOG_METRICS = {
        "count": "Count(*)",
        "duration_sum": "Sum(response_duration)",
        "duration_max": "Max(response_duration)",
        "request_length_sum": "Sum(request_length)",
        "response_length_sum": "Sum(response_length)",
    }
    # le bounds for flow_log_duration_bucket, seconds (prom convention)
    LE_BOUNDS_S = [0.001, 0.005, 0.01, 0.025, 0.05, 0.1, 0.2, 0.3, 0.5,
                   1.0, 5.0]

    def __init__(self, app_rows_fn, net_rows_fn=None, raw_sources=None,
                 sql_engine=None):
        """rows_fns return the 1s rollup dict-rows; raw_sources are
        callables (metric_name, matchers) -> series list (e.g. the
        prometheus remote-write store); sql_engine routes flow_log_*
        metrics through the GPU query engine."""
        self.sources = {"application": app_rows_fn}
        if net_rows_fn is not None:
            self.sources["network"] = net_rows_fn
        self.raw_sources = list(raw_sources or [])
        self.sql_engine = sql_engine
        self._by_hint: List[str] = []

    # ----------------------------------------------- flow_log SQL offload
    def _flow_log_where(self, matchers) -> str:
        conds = []
        for lname, op, lval in matchers:
            if lname == "le":
                continue
            if op == "=~":
                raise PromError("regex matchers are not supported for "
                                "flow_log_* metrics")
            sqlop = "=" if op == "=" else "!="
            lit = lval if lval.lstrip("-").isdigit() else f"'{lval}'"
            conds.append(f"{lname} {sqlop} {lit}")
        return (" WHERE " + " AND ".join(conds)) if conds else ""

    def _sql_series(self, agg_expr: str, matchers, name: str,
                    extra_where: str = "",
                    extra_labels: Optional[Dict] = None) -> List[Dict]:
        by = [b for b in self._by_hint if b != "le"]
        cols = ", ".join(["time(1) AS time"] + by)
        where = self._flow_log_where(matchers)
        if extra_where:
            where = (where + " AND " if where else " WHERE ") + extra_where
        group = ", ".join(["time(1)"] + by)
        sql = (f"SELECT {cols}, {agg_expr} AS v FROM l7_flow_log"
               f"{where} GROUP BY {group}")
        res = self.sql_engine.query(sql)
        cols_out = res["columns"]
        vi = cols_out.index("v")
        ti = cols_out.index("time")
        series: Dict[tuple, Dict] = {}
        for row in res["values"]:
            labels = {cols_out[i]: str(row[i])
                      for i in range(len(cols_out)) if i not in (vi, ti)}
            if extra_labels:
                labels.update(extra_labels)
            key = tuple(sorted(labels.items()))
            s = series.setdefault(key, {"metric": dict(labels,
                                                       __name__=name),
                                        "samples": {}})
            t = int(row[ti])
            s["samples"][t] = s["samples"].get(t, 0) + (row[vi] or 0)
        return list(series.values())

    def _series_flow_log(self, name: str, matchers) -> List[Dict]:
        field = name[len("flow_log_"):]
        if field == "duration_bucket":
            out = []
            for le in self.LE_BOUNDS_S:
                out.extend(self._sql_series(
                    "Count(*)", matchers, name,
                    extra_where=f"response_duration <= {int(le * 1e6)}",
                    extra_labels={"le": repr(le)}))
            out.extend(self._sql_series(
                "Count(*)", matchers, name, extra_labels={"le": "+Inf"}))
            return out
        agg = self.FLOW_LOG_METRICS.get(field)
        if agg is None:
            raise PromError(f"unknown flow_log metric {field!r}")
        return self._sql_series(agg, matchers, name)

    # -------------------------------------------------------- series fetch
    def _series(self, name: str,
                matchers: List[Tuple[str, str, str]]) -> List[Dict]:
        if name.startswith("flow_log_") and self.sql_engine is not None:
            return self._series_flow_log(name, matchers)
        for prefix, fn in self.sources.items():
            if name.startswith(prefix + "_"):
                field = name[len(prefix) + 1:]
                rows = fn()
                break
        else:
            for raw in self.raw_sources:
                series = raw(name, matchers)
                if series:
                    return series
            raise PromError(f"unknown metric {name!r}")
        series: Dict[tuple, Dict] = {}
        for r in rows:
            if field not in r:
                raise PromError(f"unknown field {field!r} for {name}")
            labels = {k: str(v) for k, v in r.items()
                      if k not in ("time", field) and not isinstance(v, float)}
            ok = True
            for lname, op, lval in matchers:
                got = labels.get(lname, "")
                if op == "=" and got != lval:
                    ok = False
                elif op == "!=" and got == lval:
                    ok = False
                elif op == "=~" and not re.fullmatch(lval, got):
                    ok = False
                if not ok:
                    break
            if not ok:
                continue
            key = tuple(sorted(labels.items()))
            s = series.setdefault(key, {"metric": dict(labels, __name__=name),
                                        "samples": {}})
            t = r["time"]
            s["samples"][t] = s["samples"].get(t, 0) + r[field]
        return list(series.values())

    # -------------------------------------------------------- evaluation
    @staticmethod
    def _split_binop(expr: str):
        """Find a top-level binary operator (lowest precedence first:
        +,- then *,/). Returns (lhs, op, rhs) or None."""
        for ops in ("+-", "*/"):
            depth = 0
            in_str = False
            for i in range(len(expr) - 1, 0, -1):
                c = expr[i]
                if c == '"':
                    in_str = not in_str
                elif in_str:
                    continue
                elif c == ")":
                    depth += 1
                elif c == "(":
                    depth -= 1
                elif depth == 0 and c in ops:
                    # don't split inside a range selector like [1m]
                    lhs, rhs = expr[:i], expr[i + 1:]
                    if not lhs.strip() or not rhs.strip():
                        continue
                    return lhs, c, rhs
        return None

    def _eval_at(self, expr: str, t: int) -> List[Dict]:
        """Evaluate expr at instant t -> [{metric, value}]."""
        expr = expr.strip()
        # histogram_quantile(q, expr): prometheus-convention le-bucket
        # interpolation over cumulative counts
        m = re.match(r"^histogram_quantile\s*\(\s*([0-9.]+)\s*,(.*)\)\s*$",
                     expr, re.DOTALL)
        if m:
            qq = float(m.group(1))
            inner = self._eval_at(m.group(2), t)
            groups: Dict[tuple, list] = {}
            for s_ in inner:
                le = s_["metric"].get("le")
                if le is None:
                    continue
                key = tuple(sorted((k, v) for k, v in s_["metric"].items()
                                   if k not in ("le", "__name__")))
                bound = float("inf") if le in ("+Inf", "Inf") else float(le)
                groups.setdefault(key, []).append((bound, s_["value"]))
            out = []
            for key, buckets in groups.items():
                buckets.sort()
                total = buckets[-1][1]
                if total <= 0:
                    continue
                target = qq * total
                prev_b, prev_c = 0.0, 0.0
                val = buckets[-1][0]
                for bound, cum in buckets:
                    if cum >= target:
                        if bound == float("inf"):
                            val = prev_b
                        else:
                            frac = (target - prev_c) / max(cum - prev_c,
                                                           1e-12)
                            val = prev_b + (bound - prev_b) * frac
                        break
                    prev_b, prev_c = bound, cum
                out.append({"metric": dict(key), "value": val})
            return out
        # topk(k, expr) / bottomk(k, expr)
        m = re.match(r"^(topk|bottomk)\s*\(\s*(\d+)\s*,(.*)\)\s*$", expr,
                     re.DOTALL)
        if m:
            k = int(m.group(2))
            inner = self._eval_at(m.group(3), t)
            inner.sort(key=lambda s_: s_["value"],
                       reverse=m.group(1) == "topk")
            return inner[:k]
        # parenthesized sub-expression
        if expr.startswith("(") :
            inner, tail = _balanced(expr[expr.index("("):])
            if inner is not None and not tail.strip():
                return self._eval_at(inner, t)
        sp = self._split_binop(expr)
        if sp is not None:
            lhs, op, rhs = sp
            try:
                rscalar = float(rhs.strip())
                left = self._eval_at(lhs, t)
                right = None
            except ValueError:
                right = self._eval_at(rhs, t)
                left = self._eval_at(lhs, t)
                rscalar = None
            out = []
            if rscalar is not None:
                for s in left:
                    out.append({"metric": s["metric"],
                                "value": _apply_op(s["value"], rscalar,
                                                   op)})
                return out
            # vector/vector: match on identical label sets (__name__
            # dropped, standard PromQL matching)
            def key(s):
                return tuple(sorted((k, v) for k, v in s["metric"].items()
                                    if k != "__name__"))
            rmap = {key(s): s["value"] for s in right}
            for s in left:
                k = key(s)
                if k in rmap:
                    out.append({"metric": {kk: vv for kk, vv in
                                           s["metric"].items()
                                           if kk != "__name__"},
                                "value": _apply_op(s["value"], rmap[k],
                                                   op)})
            return out
        agg = _match_agg(expr)
        if agg:
            func, by, inner_expr = agg
            by_labels = [x.strip() for x in by.split(",")] if by else []
            # push the by-labels down so flow_log_* selectors GROUP BY
            # exactly these dimensions on the GPU (converters.go analog)
            saved_hint = self._by_hint
            self._by_hint = by_labels
            try:
                inner = self._eval_at(inner_expr, t)
            finally:
                self._by_hint = saved_hint
            groups: Dict[tuple, List[float]] = {}
            metas: Dict[tuple, Dict] = {}
            for s in inner:
                labels = {k: v for k, v in s["metric"].items()
                          if k in by_labels}
                key = tuple(sorted(labels.items()))
                groups.setdefault(key, []).append(s["value"])
                metas[key] = labels
            out = []
            for key, vals in groups.items():
                if func == "sum":
                    v = sum(vals)
                elif func == "avg":
                    v = sum(vals) / len(vals)
                elif func == "min":
                    v = min(vals)
                elif func == "max":
                    v = max(vals)
                else:
                    v = len(vals)
                out.append({"metric": metas[key], "value": v})
            return out
        m = _FN_RE.match(expr)
        if m:
            fn = m.group("fn")
            sel = _SEL_RE.match(m.group("inner"))
            if not sel or not sel.group("range"):
                raise PromError(f"{fn}() needs a range selector")
            w = int(sel.group("range")) * _UNIT[sel.group("runit")]
            series = self._series(sel.group("name"),
                                  _parse_matchers(sel.group("matchers")))
            out = []
            for s in series:
                pts = sorted((ts, v) for ts, v in s["samples"].items()
                             if t - w < ts <= t)
                window = [v for _, v in pts]
                if not window:
                    continue
                if s.get("kind") == "counter" and fn in (
                        "rate", "irate", "increase"):
                    # cumulative counter semantics: last-first over the
                    # window (reset -> restart from the last value)
                    if len(pts) < 2:
                        continue
                    if fn == "irate":
                        (t0, v0), (t1, v1) = pts[-2], pts[-1]
                    else:
                        (t0, v0), (t1, v1) = pts[0], pts[-1]
                    delta = v1 - v0 if v1 >= v0 else v1
                    span = max(t1 - t0, 1)
                    v = delta / span if fn in ("rate", "irate") \
                        else float(delta)
                elif fn in ("rate", "irate"):
                    v = sum(window) / w
                elif fn in ("increase", "sum_over_time"):
                    v = float(sum(window))
                elif fn == "avg_over_time":
                    v = sum(window) / len(window)
                else:  # max_over_time
                    v = float(max(window))
                metric = {k: v2 for k, v2 in s["metric"].items()
                          if k != "__name__"}
                out.append({"metric": metric, "value": v})
            return out
        sel = _SEL_RE.match(expr)
        if sel and not sel.group("range"):
            series = self._series(sel.group("name"),
                                  _parse_matchers(sel.group("matchers")))
            out = []
            for s in series:
                recent = [(ts, v) for ts, v in s["samples"].items()
                          if t - self.DEFAULT_LOOKBACK < ts <= t]
                if not recent:
                    continue
                out.append({"metric": s["metric"],
                            "value": float(max(recent)[1])})
            return out
        raise PromError(f"cannot parse {expr!r}")

    # -------------------------------------------------------- public API
    def instant(self, query: str, t: Optional[int] = None) -> Dict:
        t = int(t if t is not None else _time.time())
        res = self._eval_at(query, t)
        return {"status": "success",
                "data": {"resultType": "vector",
                         "result": [{"metric": s["metric"],
                                     "value": [t, str(s["value"])]}
                                    for s in res]}}

    def range_query(self, query: str, start: int, end: int,
                    step: int) -> Dict:
        series: Dict[tuple, Dict] = {}
        t = start
        while t <= end:
            for s in self._eval_at(query, t):
                key = tuple(sorted(s["metric"].items()))
                e = series.setdefault(key, {"metric": s["metric"],
                                            "values": []})
                e["values"].append([t, str(s["value"])])
            t += step
        return {"status": "success",
                "data": {"resultType": "matrix",
                         "result": list(series.values())}}

    def label_values(self, label: str) -> Dict:
        vals = set()
        for fn in self.sources.values():
            for r in fn():
                if label in r:
                    vals.add(str(r[label]))
        return {"status": "success", "data": sorted(vals)}

    def register(self, app) -> None:
        @app.get("/prom/api/v1/query")
        @app.post("/prom/api/v1/query")
        async def prom_query(request: Request):
            params = dict(request.query_params)
            q = params.get("query")
            t = params.get("time")
            try:
                return self.instant(q, int(float(t)) if t else None)
            except PromError as e:
                return {"status": "error", "errorType": "bad_data",
                        "error": str(e)}

        @app.get("/prom/api/v1/query_range")
        @app.post("/prom/api/v1/query_range")
        async def prom_range(request: Request):
            params = dict(request.query_params)
            try:
                return self.range_query(
                    params["query"], int(float(params["start"])),
                    int(float(params["end"])),
                    int(float(params.get("step", "60"))))
            except PromError as e:
                return {"status": "error", "errorType": "bad_data",
                        "error": str(e)}

        @app.get("/prom/api/v1/label/{label}/values")
        async def prom_label_values(label: str):
            return self.label_values(label)

        @app.get("/prom/api/v1/series")
        async def prom_series(request: Request):
            m = request.query_params.get("match[]", "")
            sel = _SEL_RE.match(m or "")
            if not sel:
                return {"status": "error", "errorType": "bad_data",
                        "error": "bad match[]"}
            try:
                series = self._series(sel.group("name"),
                                      _parse_matchers(sel.group("matchers")))
            except PromError as e:
                return {"status": "error", "errorType": "bad_data",
                        "error": str(e)}
            return {"status": "success",
                    "data": [s["metric"] for s in series]}
