"""Grafana Tempo API over l7_flow_log (reference: server/querier/tempo/
tempo.go, routes querier/router/query.go:33-37).

GET /api/traces/{trace_id}   -> Tempo-style trace JSON (resource spans)
GET /api/search?tags=...     -> recent trace summaries
GET /api/echo                -> liveness (Grafana datasource check)
"""
from __future__ import annotations

from typing import Dict, List


class TempoApp:
    def __init__(self, engine):
        self.engine = engine

    def trace_by_id(self, trace_id: str) -> Dict:
        r = self.engine.query(
            "SELECT trace_id, span_id, parent_span_id, request_resource, "
            "request_domain, service_name, start_time, end_time, "
            "response_status, response_code, l7_protocol, ip4_0, ip4_1 "
            f"FROM l7_flow_log WHERE trace_id = '{trace_id}' LIMIT 10000")
        cols = r["columns"]
        spans: List[Dict] = []
        for row in r["values"]:
            d = dict(zip(cols, row))
            spans.append({
                "traceID": d["trace_id"],
                "spanID": d["span_id"],
                "parentSpanID": d.get("parent_span_id") or "",
                "operationName": d.get("request_resource") or "",
                "startTimeUnixNano": str(d["start_time"]),
                "durationNanos": str(max(d["end_time"] - d["start_time"], 0)),
                "serviceName": d.get("service_name") or "",
                "tags": [
                    {"key": "l7_protocol", "value": str(d["l7_protocol"])},
                    {"key": "response_status", "value": str(d["response_status"])},
                    {"key": "response_code", "value": str(d["response_code"])},
                    {"key": "client_ip", "value": str(d["ip4_0"])},
                    {"key": "server_ip", "value": str(d["ip4_1"])},
                ],
            })
        # Tempo v1 response shape: batches grouped by service
        by_service: Dict[str, List[Dict]] = {}
        for s in spans:
            by_service.setdefault(s["serviceName"], []).append(s)
        batches = []
        for svc, ss in by_service.items():
            batches.append({
                "resource": {"attributes": [
                    {"key": "service.name", "value": {"stringValue": svc}}]},
                "scopeSpans": [{"spans": [
                    {k: v for k, v in s.items() if k != "serviceName"}
                    for s in ss]}],
            })
        return {"batches": batches, "spanCount": len(spans)}

    def search(self, limit: int = 20, min_duration_us: int = 0) -> Dict:
        sql = ("SELECT trace_id, service_name, request_resource, start_time, "
               "end_time FROM l7_flow_log")
        if min_duration_us:
            sql += f" WHERE response_duration >= {min_duration_us}"
        sql += f" LIMIT {limit}"
        r = self.engine.query(sql)
        cols = r["columns"]
        traces = []
        seen = set()
        for row in r["values"]:
            d = dict(zip(cols, row))
            tid = d["trace_id"]
            if not tid or tid in seen:
                continue
            seen.add(tid)
            traces.append({
                "traceID": tid,
                "rootServiceName": d.get("service_name") or "",
                "rootTraceName": d.get("request_resource") or "",
                "startTimeUnixNano": str(d["start_time"]),
                "durationMs": max(d["end_time"] - d["start_time"], 0) // 10**6,
            })
        return {"traces": traces}

    def register(self, app) -> None:
        @app.get("/api/echo")
        def echo():
            return "echo"

        @app.get("/api/traces/{trace_id}")
        def get_trace(trace_id: str):
            return self.trace_by_id(trace_id)

        @app.get("/api/search")
        def search(limit: int = 20, minDuration: int = 0):
            return self.search_route(limit, minDuration)

        # keep bound method accessible for the closure above
        self.search_route = lambda limit, mind: self.search(limit, mind)
