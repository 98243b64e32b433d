"""db_descriptions: per-table tag + metric metadata for discovery APIs.

Reference counterpart: server/querier/db_descriptions/clickhouse/
(203 data files: every tag and metric per table with display names,
units, types and descriptions, driving `show tags/metrics`). Here the
entries are GENERATED from the live tag maps and rollup table defs, then
overlaid with curated display/unit/description text — so the surface can
never drift from what is actually queryable (the reference's files go
stale instead).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

# curated field metadata: name -> (display, unit, description)
FIELD_META: Dict[str, Tuple[str, str, str]] = {
    "time": ("Time", "s", "Row timestamp, epoch seconds"),
    "start_time": ("Start Time", "ns", "Flow/span start timestamp"),
    "end_time": ("End Time", "ns", "Flow/span end timestamp"),
    "flow_id": ("Flow ID", "", "Unique flow identifier from the agent"),
    "vtap_id": ("Agent", "", "Collecting agent (vtap) id"),
    "agent_id": ("Agent", "", "Collecting agent (vtap) id"),
    "ip4_0": ("Client IPv4", "", "Client (initiator) IPv4 address"),
    "ip4_1": ("Server IPv4", "", "Server (responder) IPv4 address"),
    "ip6_0": ("Client IPv6", "", "Client IPv6 address (empty for v4)"),
    "ip6_1": ("Server IPv6", "", "Server IPv6 address (empty for v4)"),
    "ip_0": ("Client IP", "", "Client endpoint address"),
    "ip_1": ("Server IP", "", "Server endpoint address"),
    "ip": ("IP", "", "Endpoint address"),
    "is_ipv6": ("Is IPv6", "", "1 when the flow is IPv6"),
    "l3_epc_id_0": ("Client EPC", "", "Client L3 endpoint-collection id"),
    "l3_epc_id_1": ("Server EPC", "", "Server L3 endpoint-collection id"),
    "l3_epc_id": ("EPC", "", "L3 endpoint-collection id"),
    "client_port": ("Client Port", "", "Client TCP/UDP port"),
    "server_port": ("Server Port", "", "Server TCP/UDP port"),
    "protocol": ("L4 Protocol", "", "IP protocol number (6=TCP, 17=UDP)"),
    "l7_protocol": ("L7 Protocol", "", "Application protocol (HTTP, "
                    "MySQL, Redis, Kafka, ...)"),
    "l7_protocol_str": ("L7 Protocol Name", "", "Application protocol "
                        "display name"),
    "tap_side": ("Tap Side", "", "Observation side (client/server)"),
    "tap_type": ("Tap Type", "", "Capture network type"),
    "type": ("Message Type", "", "0=request 1=response 2=session"),
    "request_type": ("Request Type", "", "Method/command of the request"),
    "request_domain": ("Request Domain", "", "Host/database/topic the "
                       "request addresses"),
    "request_resource": ("Request Resource", "", "Path/table/queue of "
                         "the request"),
    "endpoint": ("Endpoint", "", "Normalized request endpoint"),
    "request_id": ("Request ID", "", "Protocol-level request id/stream id"),
    "response_status": ("Response Status", "", "0=ok 3=server error "
                        "4=client error"),
    "response_code": ("Response Code", "", "Protocol response code"),
    "response_result": ("Response Result", "", "Response payload excerpt"),
    "exception_desc": ("Exception", "", "Error/exception description"),
    "version": ("Protocol Version", "", "Application protocol version"),
    "trace_id": ("Trace ID", "", "Distributed trace id (OTel)"),
    "span_id": ("Span ID", "", "Span id (OTel)"),
    "parent_span_id": ("Parent Span ID", "", "Parent span id"),
    "x_request_id_0": ("X-Request-ID (rx)", "", "Ingress x-request-id"),
    "x_request_id_1": ("X-Request-ID (tx)", "", "Egress x-request-id"),
    "http_user_agent": ("User Agent", "", "HTTP User-Agent header"),
    "http_referer": ("Referer", "", "HTTP Referer header"),
    "service_name": ("Service", "", "OTel service.name resource"),
    "app_service": ("App Service", "", "Application service name"),
    "process_id_0": ("Client PID", "", "Client-side process id"),
    "process_id_1": ("Server PID", "", "Server-side process id"),
    "process_kname_0": ("Client Kernel Thread", "", "Client kthread name"),
    "process_kname_1": ("Server Kernel Thread", "", "Server kthread name"),
    "gprocess_id_0": ("Client GProcess", "", "Global process id (client)"),
    "gprocess_id_1": ("Server GProcess", "", "Global process id (server)"),
    "syscall_trace_id_request": ("Syscall Trace ID (req)", "",
                                 "eBPF syscall join key of the request"),
    "syscall_trace_id_response": ("Syscall Trace ID (resp)", "",
                                  "eBPF syscall join key of the response"),
    "acl_gid": ("ACL Group", "", "Matched policy ACL group id"),
    "tap_port": ("Tap Port", "", "Capture port identifier"),
    "biz_type": ("Business Type", "", "Tenant-defined business type"),
    # metrics
    "log_count": ("Log Count", "count", "Rows matching the query"),
    "response_duration": ("Response Duration", "us",
                          "Request->response latency"),
    "request_length": ("Request Length", "byte", "Request payload bytes"),
    "response_length": ("Response Length", "byte",
                        "Response payload bytes"),
    "captured_request_byte": ("Captured Request Bytes", "byte", ""),
    "captured_response_byte": ("Captured Response Bytes", "byte", ""),
    "byte_tx": ("Bytes TX", "byte", "Client->server bytes"),
    "byte_rx": ("Bytes RX", "byte", "Server->client bytes"),
    "packet_tx": ("Packets TX", "count", "Client->server packets"),
    "packet_rx": ("Packets RX", "count", "Server->client packets"),
    "total_byte_tx": ("Total Bytes TX", "byte", ""),
    "total_byte_rx": ("Total Bytes RX", "byte", ""),
    "new_flow": ("New Flows", "count", "Flows opened in the window"),
    "closed_flow": ("Closed Flows", "count", "Flows closed in the window"),
    "request": ("Requests", "count", "L7 requests"),
    "response": ("Responses", "count", "L7 responses"),
    "client_error": ("Client Errors", "count", "4xx-class responses"),
    "server_error": ("Server Errors", "count", "5xx-class responses"),
    "rrt_sum": ("RRT Sum", "us", "Sum of request-response times"),
    "rrt_count": ("RRT Count", "count", "Samples in rrt_sum"),
    "rrt_max": ("RRT Max", "us", "Slowest request-response time"),
    "rtt": ("TCP RTT", "us", "Handshake round-trip time"),
    "rtt_sum": ("RTT Sum", "us", ""),
    "rtt_count": ("RTT Count", "count", ""),
    "rtt_max": ("RTT Max", "us", ""),
    "srt_sum": ("SRT Sum", "us", "System response time sum (data->ACK)"),
    "srt_count": ("SRT Count", "count", ""),
    "srt_max": ("SRT Max", "us", ""),
    "art_sum": ("ART Sum", "us", "Application response time sum"),
    "art_count": ("ART Count", "count", ""),
    "art_max": ("ART Max", "us", ""),
    "cit_sum": ("CIT Sum", "us", "Client idle time sum"),
    "cit_count": ("CIT Count", "count", ""),
    "cit_max": ("CIT Max", "us", ""),
    "retrans": ("Retransmissions", "count", ""),
    "retrans_tx": ("Retrans TX", "count", ""),
    "retrans_rx": ("Retrans RX", "count", ""),
    "retrans_total": ("Retrans Total", "count", ""),
    "zero_win_tx": ("Zero Window TX", "count", ""),
    "zero_win_rx": ("Zero Window RX", "count", ""),
    "syn_count": ("SYN Count", "count", ""),
    "synack_count": ("SYN-ACK Count", "count", ""),
    "duration": ("Duration", "ns", "Flow duration"),
    "close_type": ("Close Type", "", "How the flow ended (1=fin 2=rst "
                   "3=timeout)"),
    "signal_source": ("Signal Source", "", "0=packet 3=eBPF"),
    "l7_request": ("L7 Requests", "count", ""),
    "l7_response": ("L7 Responses", "count", ""),
}

# KG universal tags share shapes per side
for side, side_d in ((0, "Client"), (1, "Server")):
    for kname, disp in [
        ("pod_id", "Pod"), ("pod_node_id", "Pod Node"),
        ("pod_ns_id", "Namespace"), ("pod_group_id", "Workload"),
        ("pod_cluster_id", "Cluster"), ("l3_device_type", "Device Type"),
        ("l3_device_id", "Device"), ("subnet_id", "Subnet"),
        ("host_id", "Host"), ("az_id", "Availability Zone"),
        ("service_id", "Service"), ("gprocess_id", "Global Process"),
    ]:
        FIELD_META[f"{kname}_{side}"] = (
            f"{side_d} {disp}", "",
            f"{side_d}-side {disp.lower()} resolved from the platform "
            f"KnowledgeGraph at query time")


def describe(name: str, kind: str = "tag") -> Dict[str, str]:
    disp, unit, desc = FIELD_META.get(name, (name, "", ""))
    return {"name": name, "display_name": disp, "unit": unit,
            "description": desc, "type": kind}


def table_descriptions(engine) -> Dict[str, Dict[str, List[Dict]]]:
    """Full catalog: table -> {tags: [...], metrics: [...]} generated
    from the engine's live tag maps + rollup table definitions."""
    from .tags import L7_TAGS, L7_METRICS, L4_TAGS, L4_METRICS
    out: Dict[str, Dict[str, List[Dict]]] = {}
    out["l7_flow_log"] = {
        "tags": [describe(n) for n in sorted(L7_TAGS)],
        "metrics": [describe(n, "metric") for n in sorted(L7_METRICS)],
    }
    out["l4_flow_log"] = {
        "tags": [describe(n) for n in sorted(L4_TAGS)],
        "metrics": [describe(n, "metric") for n in sorted(L4_METRICS)],
    }
    for pipe in (engine.pipe, engine.l4):
        if pipe is None or not hasattr(pipe, "rollups"):
            continue
        for tname, table in pipe.rollups.tables.items():
            names = table.td.out_names or table.td.keys
            out[tname] = {
                "tags": [describe("time")] + [describe(n) for n in names],
                "metrics": [describe(f, "metric") for f in table.fields],
            }
    return out
