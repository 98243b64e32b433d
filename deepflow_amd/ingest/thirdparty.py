"""Third-party APM trace adapters: SkyWalking + Datadog -> L7 span records.

Reference counterparts: the ingester's SkyWalking/Datadog loggers (decoder
msg types 19/20 carrying ThirdPartyTrace, flow_log.proto:317) and the
querier tracing-adapter that pulls SkyWalking segments
(querier/app/tracing-adapter/service/skywalking.go). Both convert foreign
span models into the same AppProtoLogsData wire form and ride the normal
GPU span pipeline.
"""
from __future__ import annotations

import json
from typing import Dict, List

from ..wire import pb, flow_log, framing
from ..wire.const_enums import (L7_PROTOCOL_HTTP_1, L7_PROTOCOL_GRPC,
                                L7_PROTOCOL_MYSQL, L7_PROTOCOL_UNKNOWN)

THIRD_PARTY_TRACE = {
    1: ("data", 'b'),
    2: ("peer_ip", 'b'),
    3: ("uri", 's'),
    4: ("extend_keys", '*s'),
    5: ("extend_values", '*s'),
}

# SkyWalking spanLayer -> l7 protocol
_SW_LAYER = {"Http": L7_PROTOCOL_HTTP_1, "RPCFramework": L7_PROTOCOL_GRPC,
             "Database": L7_PROTOCOL_MYSQL}


def skywalking_segment_to_l7(segment: Dict) -> List[Dict]:
    """SkyWalking v3 segment object -> AppProtoLogsData dicts."""
    out = []
    trace_id = segment.get("traceId", "")
    service = segment.get("service", "")
    for span in segment.get("spans", []):
        tags = {t.get("key"): t.get("value")
                for t in span.get("tags", [])}
        layer = span.get("spanLayer", "")
        start_ms = span.get("startTime", 0)
        end_ms = span.get("endTime", start_ms)
        span_id = f"{segment.get('traceSegmentId', '')}-{span.get('spanId', 0)}"
        parent = span.get("parentSpanId", -1)
        parent_span_id = "" if parent in (-1, None) else \
            f"{segment.get('traceSegmentId', '')}-{parent}"
        refs = span.get("refs", [])
        if parent in (-1, None) and refs:
            parent_span_id = f"{refs[0].get('parentTraceSegmentId', '')}-" \
                             f"{refs[0].get('parentSpanId', 0)}"
        rec = {
            "base": {
                "start_time": start_ms * 1_000_000,
                "end_time": end_ms * 1_000_000,
                "tap_side": 1 if span.get("spanType") == "Exit" else 2,
                "head": {"proto": _SW_LAYER.get(layer, L7_PROTOCOL_UNKNOWN),
                         "msg_type": 2,
                         "rrt": max(end_ms - start_ms, 0) * 1000},
            },
            "req": {
                "req_type": tags.get("http.method", ""),
                "domain": span.get("peer", ""),
                "resource": span.get("operationName", ""),
                "endpoint": span.get("operationName", ""),
            },
            "resp": {
                "status": 3 if span.get("isError") else 0,
                "code": int(tags.get("http.status_code") or
                            tags.get("status_code") or 0),
            },
            "trace_info": {"trace_id": trace_id, "span_id": span_id,
                           "parent_span_id": parent_span_id},
            "ext_info": {
                "service_name": service,
                "attribute_names": list(tags.keys()),
                "attribute_values": [str(v) for v in tags.values()],
            },
        }
        out.append(rec)
    return out


def datadog_traces_to_l7(traces) -> List[Dict]:
    """Datadog agent trace payload (list of traces, each a list of spans)."""
    out = []
    for trace in traces:
        for span in trace:
            meta = span.get("meta", {})
            rec = {
                "base": {
                    "start_time": span.get("start", 0),
                    "end_time": span.get("start", 0) +
                    span.get("duration", 0),
                    "tap_side": 1,
                    "head": {
                        "proto": L7_PROTOCOL_HTTP_1
                        if span.get("type") == "web" else
                        L7_PROTOCOL_UNKNOWN,
                        "msg_type": 2,
                        "rrt": span.get("duration", 0) // 1000,
                    },
                },
                "req": {
                    "req_type": meta.get("http.method", ""),
                    "domain": meta.get("http.host", ""),
                    "resource": span.get("resource", span.get("name", "")),
                    "endpoint": span.get("name", ""),
                },
                "resp": {"status": 3 if span.get("error") else 0,
                         "code": int(meta.get("http.status_code") or 0)},
                "trace_info": {
                    "trace_id": "%032x" % span.get("trace_id", 0),
                    "span_id": "%016x" % span.get("span_id", 0),
                    "parent_span_id": "%016x" % span.get("parent_id", 0)
                    if span.get("parent_id") else "",
                },
                "ext_info": {
                    "service_name": span.get("service", ""),
                    "attribute_names": list(meta.keys()),
                    "attribute_values": [str(v) for v in meta.values()],
                },
            }
            out.append(rec)
    return out


def third_party_frame_to_l7_payload(payload: bytes, kind: str) -> bytes:
    """MSG_SKYWALKING / MSG_DATADOG frame -> l7 record payload. The frame
    carries a ThirdPartyTrace whose data is the foreign JSON."""
    tpt = pb.decode(payload, THIRD_PARTY_TRACE)
    data = json.loads(tpt.get("data", b"{}").decode("utf-8", "replace"))
    if kind == "skywalking":
        segments = data if isinstance(data, list) else [data]
        recs = []
        for seg in segments:
            recs.extend(skywalking_segment_to_l7(seg))
    else:
        recs = datadog_traces_to_l7(data)
    return framing.pack_records(
        [pb.encode(r, flow_log.APP_PROTO_LOGS_DATA) for r in recs])


class TracingAdapter:
    """Querier-side adapter (reference app/tracing-adapter): pulls external
    APM traces into the DeepFlow trace view on demand."""

    def __init__(self, ingest_fn):
        self.ingest = ingest_fn

    def import_skywalking_segments(self, segments: List[Dict]) -> int:
        recs = []
        for seg in segments:
            recs.extend(skywalking_segment_to_l7(seg))
        payload = framing.pack_records(
            [pb.encode(r, flow_log.APP_PROTO_LOGS_DATA) for r in recs])
        self.ingest(payload)
        return len(recs)
