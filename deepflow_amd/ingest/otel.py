"""OTLP trace ingestion: TracesData -> AppProtoLogsData records.

The semantic mapping mirrors the reference's OTelTracesDataToL7FlowLogs
(server/ingester/flow_log/log_data/otel_import.go:39-343; SURVEY.md
appendix E): span kind -> tap_side, well-known http.*/db.*/rpc.* attributes
-> typed L7 fields, remaining attributes -> attribute_names/values, resource
service.name -> service_name. The converted records then ride the normal
GPU span pipeline (decode runs on the AppProtoLogsData wire form).
"""
from __future__ import annotations

import zlib
from typing import Dict, List

from ..wire import pb, otlp, flow_log, framing
from ..wire.const_enums import (L7_PROTOCOL_HTTP_1, L7_PROTOCOL_GRPC,
                                L7_PROTOCOL_MYSQL, L7_PROTOCOL_REDIS,
                                L7_PROTOCOL_UNKNOWN)

_DB_PROTO = {"mysql": L7_PROTOCOL_MYSQL, "redis": L7_PROTOCOL_REDIS}


def _attr_val(v: Dict) -> str:
    if not v:
        return ""
    if "string_value" in v:
        return v["string_value"]
    if "int_value" in v:
        return str(v["int_value"])
    if "bool_value" in v:
        return "true" if v["bool_value"] else "false"
    if "double_value" in v:
        return str(v["double_value"])
    return ""


def span_to_l7(span: Dict, resource_attrs: Dict[str, str]) -> Dict:
    attrs = {kv.get("key", ""): _attr_val(kv.get("value", {}))
             for kv in span.get("attributes", [])}
    kind = span.get("kind", 0)
    tap_side = 1 if kind == otlp.SPAN_KIND_CLIENT else \
        (2 if kind == otlp.SPAN_KIND_SERVER else 0)
    proto = L7_PROTOCOL_UNKNOWN
    req_type, domain, resource, endpoint = "", "", "", ""
    code = 0
    if "http.method" in attrs or "http.request.method" in attrs:
        proto = L7_PROTOCOL_HTTP_1
        req_type = attrs.get("http.method") or attrs.get("http.request.method")
        resource = attrs.get("http.target") or attrs.get("url.path") or \
            attrs.get("http.url", "")
        domain = attrs.get("http.host") or attrs.get("server.address", "")
        endpoint = span.get("name", "")
        code = int(attrs.get("http.status_code") or
                   attrs.get("http.response.status_code") or 0)
    elif "rpc.system" in attrs:
        proto = L7_PROTOCOL_GRPC
        req_type = attrs.get("rpc.method", "")
        domain = attrs.get("rpc.service", "")
        resource = span.get("name", "")
        endpoint = resource
        code = int(attrs.get("rpc.grpc.status_code") or 0)
    elif "db.system" in attrs:
        proto = _DB_PROTO.get(attrs.get("db.system", ""), L7_PROTOCOL_MYSQL)
        req_type = attrs.get("db.operation", "")
        domain = attrs.get("db.name", "")
        resource = attrs.get("db.statement", "") or span.get("name", "")
        endpoint = req_type
    else:
        resource = span.get("name", "")
        endpoint = resource
    status_code = span.get("status", {}).get("code", 0)
    status = 0 if status_code != 2 else (4 if 400 <= code < 500 else 3)
    well_known = {"http.method", "http.request.method", "http.target",
                  "url.path", "http.url", "http.host", "server.address",
                  "http.status_code", "http.response.status_code",
                  "rpc.system", "rpc.method", "rpc.service",
                  "rpc.grpc.status_code", "db.system", "db.operation",
                  "db.name", "db.statement"}
    extra = {k: v for k, v in attrs.items() if k not in well_known}
    rec = {
        "base": {
            "start_time": span.get("start_time_unix_nano", 0),
            "end_time": span.get("end_time_unix_nano", 0),
            "tap_side": tap_side,
            "head": {
                "proto": proto,
                "msg_type": 2,
                "rrt": max(span.get("end_time_unix_nano", 0) -
                           span.get("start_time_unix_nano", 0), 0) // 1000,
            },
        },
        "req": {"req_type": req_type, "domain": domain,
                "resource": resource, "endpoint": endpoint},
        "resp": {"status": status, "code": code},
        "trace_info": {
            "trace_id": span.get("trace_id", b"").hex(),
            "span_id": span.get("span_id", b"").hex(),
            "parent_span_id": span.get("parent_span_id", b"").hex(),
        },
        "ext_info": {
            "service_name": resource_attrs.get("service.name", ""),
            "attribute_names": list(extra.keys()),
            "attribute_values": list(extra.values()),
        },
    }
    return rec


def otlp_to_l7_payload(data: bytes, compressed: bool = False) -> bytes:
    """OTLP TracesData bytes -> length-prefixed AppProtoLogsData payload."""
    if compressed:
        data = zlib.decompress(data)
    td = pb.decode(data, otlp.TRACES_DATA)
    records: List[bytes] = []
    for rs in td.get("resource_spans", []):
        res_attrs = {kv.get("key", ""): _attr_val(kv.get("value", {}))
                     for kv in rs.get("resource", {}).get("attributes", [])}
        for ss in rs.get("scope_spans", []):
            for span in ss.get("spans", []):
                rec = span_to_l7(span, res_attrs)
                records.append(pb.encode(rec, flow_log.APP_PROTO_LOGS_DATA))
    return framing.pack_records(records)


def otlp_logs_to_rows(data: bytes, agent_id: int = 0) -> List[Dict]:
    """OTLP LogsData bytes -> application_log rows (reference:
    server/ingester/app_log otel log import)."""
    ld = pb.decode(data, otlp.LOGS_DATA)
    rows: List[Dict] = []
    for rl in ld.get("resource_logs", []):
        res_attrs = {kv.get("key", ""): _attr_val(kv.get("value", {}))
                     for kv in rl.get("resource", {}).get("attributes", [])}
        svc = res_attrs.get("service.name", "")
        for sl in rl.get("scope_logs", []):
            for rec in sl.get("log_records", []):
                ts = rec.get("time_unix_nano", 0) or \
                    rec.get("observed_time_unix_nano", 0)
                attrs = {kv.get("key", ""): _attr_val(kv.get("value", {}))
                         for kv in rec.get("attributes", [])}
                rows.append({
                    "time": ts // 10**9,
                    "agent_id": agent_id,
                    "log_type": "otlp",
                    # OTLP severity bands (1-4 trace .. 21-24 fatal)
                    # -> syslog levels
                    "severity": {0: 7, 1: 7, 2: 6, 3: 4, 4: 3, 5: 2}[
                        min(max(rec.get("severity_number", 9) - 1, 0) // 4,
                            5)],
                    "severity_text": rec.get("severity_text", ""),
                    "app_service": svc,
                    "trace_id": rec.get("trace_id", b"").hex(),
                    "span_id": rec.get("span_id", b"").hex(),
                    "body": _attr_val(rec.get("body", {})),
                    **{f"attr.{k}": v for k, v in attrs.items()},
                })
    return rows


def _dp_labels(res_attrs: Dict[str, str], dp: Dict) -> Dict[str, str]:
    labels = dict(res_attrs)
    for kv in dp.get("attributes", []):
        labels[kv.get("key", "")] = _attr_val(kv.get("value", {}))
    return labels


def otlp_metrics_to_samples(data: bytes) -> List[tuple]:
    """OTLP MetricsData bytes -> [(metric_name, labels, ts_ms, value)].

    Gauge/Sum map 1:1; Histogram expands to the Prometheus convention
    (<name>_count, <name>_sum, <name>_bucket{le=...}) so PromQL sees the
    same series a prometheus remote-write would produce."""
    md = pb.decode(data, otlp.METRICS_DATA)
    out: List[tuple] = []
    for rm in md.get("resource_metrics", []):
        res_attrs = {kv.get("key", ""): _attr_val(kv.get("value", {}))
                     for kv in rm.get("resource", {}).get("attributes", [])}
        res_attrs = {k: v for k, v in res_attrs.items()
                     if k in ("service.name", "host.name")}
        for sm in rm.get("scope_metrics", []):
            for m in sm.get("metrics", []):
                name = m.get("name", "")
                for dp in m.get("gauge", {}).get("data_points", []) + \
                        m.get("sum", {}).get("data_points", []):
                    v = dp.get("as_double", 0.0) or float(
                        _sfixed64(dp.get("as_int", 0)))
                    out.append((name, _dp_labels(res_attrs, dp),
                                dp.get("time_unix_nano", 0) // 10**6, v))
                for dp in m.get("histogram", {}).get("data_points", []):
                    ts = dp.get("time_unix_nano", 0) // 10**6
                    labels = _dp_labels(res_attrs, dp)
                    out.append((name + "_count", labels, ts,
                                float(dp.get("count", 0))))
                    out.append((name + "_sum", labels, ts,
                                float(dp.get("sum", 0.0))))
                    bounds = dp.get("explicit_bounds", [])
                    cum = 0
                    for bi, c in enumerate(dp.get("bucket_counts", [])):
                        cum += c
                        le = str(bounds[bi]) if bi < len(bounds) else "+Inf"
                        out.append((name + "_bucket",
                                    {**labels, "le": le}, ts, float(cum)))
    return out


def _sfixed64(v: int) -> int:
    return v - (1 << 64) if v >= (1 << 63) else v
