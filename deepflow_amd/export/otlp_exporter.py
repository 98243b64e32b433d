"""OTLP re-export of enriched flow logs.

Reference counterpart: server/ingester/exporters (otlp_export.go:709) —
re-emits stored, tag-enriched L7 spans as OTLP traces for downstream APMs.
Export pulls rows through the query engine (so SmartEncoding hydration and
KG enrichment are applied), builds TracesData protobuf, and hands bytes to
a sink (HTTP post or caller-provided)."""
from __future__ import annotations

import zlib
from typing import Callable, Dict, List, Optional

from ..wire import pb, otlp


def _kv(key: str, val) -> Dict:
    if isinstance(val, int):
        return {"key": key, "value": {"int_value": val}}
    return {"key": key, "value": {"string_value": str(val)}}


class OtlpExporter:
    COLUMNS = ("trace_id, span_id, parent_span_id, request_resource, "
               "request_domain, service_name, start_time, end_time, "
               "response_status, response_code, l7_protocol, ip4_0, ip4_1, "
               "pod_id_1, flow_id")

    def __init__(self, engine, sink: Optional[Callable[[bytes], None]] = None):
        self.engine = engine
        self.sink = sink
        self.exported = 0

    def export_where(self, where: str = "", limit: int = 10000) -> bytes:
        sql = f"SELECT {self.COLUMNS} FROM l7_flow_log"
        if where:
            sql += f" WHERE {where}"
        sql += f" LIMIT {limit}"
        r = self.engine.query(sql)
        cols = r["columns"]
        by_service: Dict[str, List[Dict]] = {}
        for row in r["values"]:
            d = dict(zip(cols, row))
            span = {
                "trace_id": bytes.fromhex(d["trace_id"])
                if d["trace_id"] and len(d["trace_id"]) == 32 else
                (d["trace_id"] or "").encode()[:16].ljust(16, b"\0"),
                "span_id": bytes.fromhex(d["span_id"])
                if d["span_id"] and len(d["span_id"]) == 16 else
                (d["span_id"] or "").encode()[:8].ljust(8, b"\0"),
                "name": d.get("request_resource") or "span",
                "kind": otlp.SPAN_KIND_CLIENT,
                "start_time_unix_nano": d["start_time"],
                "end_time_unix_nano": d["end_time"],
                "attributes": [
                    _kv("df.domain", d.get("request_domain") or ""),
                    _kv("df.response_code", d.get("response_code") or 0),
                    _kv("df.client_ip", d.get("ip4_0") or ""),
                    _kv("df.server_ip", d.get("ip4_1") or ""),
                    _kv("df.server_pod_id", d.get("pod_id_1") or 0),
                    _kv("df.flow_id", d.get("flow_id") or 0),
                ],
                "status": {"code": 2 if d.get("response_status") in
                           ("Server Error", "Client Error") else 1},
            }
            by_service.setdefault(d.get("service_name") or "unknown",
                                  []).append(span)
        td = {"resource_spans": [
            {"resource": {"attributes": [_kv("service.name", svc)]},
             "scope_spans": [{"scope": {"name": "deepflow-amd"},
                              "spans": spans}]}
            for svc, spans in by_service.items()]}
        blob = pb.encode(td, otlp.TRACES_DATA)
        self.exported += sum(len(s) for s in by_service.values())
        if self.sink:
            self.sink(blob)
        return blob

    def export_compressed(self, where: str = "", limit: int = 10000) -> bytes:
        return zlib.compress(self.export_where(where, limit))
