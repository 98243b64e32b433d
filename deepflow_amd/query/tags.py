"""Tag resolution: DF-SQL tag names -> column sources + hydration metadata.

The moral equivalent of the reference querier's tag translation layer
(server/querier/engine/clickhouse/tag/translation.go): each queryable tag
maps to a (family, idx) column source; dict-encoded tags carry their
SmartEncoding domain so filters can compile string literals to IDs and
results can be hydrated back to names.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional

from ..store import l7_schema as S
from . import spec as Q


@dataclass
class TagDef:
    name: str
    family: int
    idx: int
    # hydration: 'int' | 'dict:<domain>' | 'ip' | 'status' | 'l7proto'
    hydrate: str = "int"
    description: str = ""


def _u64(name, col, **kw):
    return TagDef(name, Q.SRC_U64, S.U64_COLS.index(col), **kw)


def _u32(name, col, **kw):
    return TagDef(name, Q.SRC_U32, S.U32_COLS.index(col), **kw)


def _u8(name, col, **kw):
    return TagDef(name, Q.SRC_U8, S.U8_COLS.index(col), **kw)


def build_l7_tags() -> Dict[str, TagDef]:
    tags: Dict[str, TagDef] = {}

    def add(t: TagDef):
        tags[t.name] = t

    add(_u64("start_time", "start_time"))
    add(_u64("end_time", "end_time"))
    add(_u64("flow_id", "flow_id"))
    add(_u64("response_duration", "rrt"))
    add(_u64("syscall_trace_id_request", "syscall_trace_id_request"))
    add(_u64("syscall_trace_id_response", "syscall_trace_id_response"))
    for name, col in [
        ("agent_id", "vtap_id"), ("vtap_id", "vtap_id"),
        ("request_id", "request_id"),
        ("response_code", "response_code"),
        ("request_length", "request_length"),
        ("response_length", "response_length"),
        ("client_port", "client_port"), ("server_port", "server_port"),
        ("req_tcp_seq", "req_tcp_seq"), ("resp_tcp_seq", "resp_tcp_seq"),
        ("captured_request_byte", "captured_request_byte"),
        ("captured_response_byte", "captured_response_byte"),
        ("biz_type", "biz_type"),
    ]:
        add(_u32(name, col))
    for side in (0, 1):
        add(_u32(f"ip4_{side}", f"ip4_{side}", hydrate="ip"))
        add(_u32(f"l3_epc_id_{side}", f"l3_epc_id_{side}"))
        add(_u32(f"process_id_{side}", f"process_id_{side}"))
    add(_u8("tap_side", "tap_side"))
    add(_u8("tap_type", "tap_type"))
    add(_u8("protocol", "protocol"))
    add(_u8("l7_protocol", "l7_protocol", hydrate="l7proto"))
    add(_u8("type", "msg_type"))
    add(_u8("response_status", "response_status", hydrate="status"))
    add(_u8("is_ipv6", "is_ipv6"))
    # dict-encoded tags
    for did_idx, (name, _, dom) in enumerate(S.DID_COLS):
        add(TagDef(name, Q.SRC_DID, did_idx, hydrate=f"dict:{dom}"))
    # aliases matching reference tag names
    tags["l7_protocol_str"] = TagDef("l7_protocol_str", Q.SRC_U8,
                                     S.U8_COLS.index("l7_protocol"),
                                     hydrate="l7proto")
    tags["app_service"] = tags["service_name"]
    # pooled string tags: filterable via GPU string-hash compare,
    # selectable from the segment pool
    for sname in ["parent_span_id", "x_request_id_0",
                  "x_request_id_1", "http_user_agent", "biz_code"]:
        add(TagDef(sname, Q.SRC_STR_HASH, S.POOL_POS[sname],
                   hydrate="strhash"))
    # binary-transcoded OTel ids: u64 columns when hex, pool fallback
    add(TagDef("trace_id", Q.SRC_TRACE128, 0, hydrate="tracebin"))
    add(TagDef("span_id", Q.SRC_TRACE128, 1, hydrate="tracebin"))
    # pooled 16-byte v6 addresses: equality filter hashes the packed
    # address bytes; select formats them back to text
    for sname in ["ip6_0", "ip6_1"]:
        add(TagDef(sname, Q.SRC_STR_HASH, S.POOL_POS[sname],
                   hydrate="ip6str"))
    # KnowledgeGraph universal tags, client (_0) / server (_1)
    for side in (0, 1):
        for j, kname in enumerate(S.KG_COLS):
            nm = f"{kname}_{side}"
            add(TagDef(nm, Q.SRC_KG, side * S.N_KG + j))
    # un-suffixed aliases -> server side (matches reference default for
    # single-ended metrics tables)
    for j, kname in enumerate(S.KG_COLS):
        tags.setdefault(kname, TagDef(kname, Q.SRC_KG, S.N_KG + j))
    # id -> display-name tags via tagrecorder name maps (reference:
    # ClickHouse dictGet on the *_map dictionaries)
    KG_NAME_MAPS = {"pod_id": "pod", "pod_node_id": "pod_node",
                    "pod_ns_id": "pod_ns", "pod_group_id": "pod_group",
                    "pod_cluster_id": "pod_cluster",
                    "service_id": "service"}
    for j, kname in enumerate(S.KG_COLS):
        mp = KG_NAME_MAPS.get(kname)
        if mp is None:
            continue
        base = kname[:-3]  # strip _id
        for side in (0, 1):
            add(TagDef(f"{base}_name_{side}", Q.SRC_KG,
                       side * S.N_KG + j, hydrate=f"kgname:{mp}"))
        tags.setdefault(f"{base}_name",
                        TagDef(f"{base}_name", Q.SRC_KG, S.N_KG + j,
                               hydrate=f"kgname:{mp}"))
    # time pseudo-tag handled by the parser (SRC_TIME_BUCKET)
    return tags


L7_TAGS = build_l7_tags()

# aggregatable metric fields (reference: db_descriptions metrics)
# db_descriptions analog: units + display info for `show metrics`
METRIC_UNITS: Dict[str, tuple] = {
    "response_duration": ("us", "Response Duration"),
    "request_length": ("byte", "Request Length"),
    "response_length": ("byte", "Response Length"),
    "captured_request_byte": ("byte", "Captured Request Bytes"),
    "captured_response_byte": ("byte", "Captured Response Bytes"),
    "log_count": ("count", "Log Count"),
    "byte_tx": ("byte", "Bytes TX"), "byte_rx": ("byte", "Bytes RX"),
    "packet_tx": ("count", "Packets TX"),
    "packet_rx": ("count", "Packets RX"),
    "rtt": ("us", "TCP Handshake RTT"),
    "srt_sum": ("us", "System Response Time Sum"),
    "art_sum": ("us", "Application Response Time Sum"),
    "retrans_tx": ("count", "Retransmits TX"),
    "retrans_rx": ("count", "Retransmits RX"),
}

L7_METRICS: Dict[str, TagDef] = {
    "response_duration": L7_TAGS["response_duration"],
    "request_length": L7_TAGS["request_length"],
    "response_length": L7_TAGS["response_length"],
    "captured_request_byte": L7_TAGS["captured_request_byte"],
    "captured_response_byte": L7_TAGS["captured_response_byte"],
    "log_count": TagDef("log_count", Q.SRC_CONST0, 0),
}


def build_l4_tags() -> Dict[str, TagDef]:
    from ..store import l4_schema as L4
    tags: Dict[str, TagDef] = {}
    for i, c in enumerate(L4.U64_COLS):
        tags[c] = TagDef(c, Q.SRC_U64, i)
    for i, c in enumerate(L4.U32_COLS):
        hyd = "ip" if c.startswith("ip4") or c.startswith("nat_real_ip") \
            else "int"
        tags[c] = TagDef(c, Q.SRC_U32, i, hydrate=hyd)
    for i, c in enumerate(L4.U8_COLS):
        hyd = "l7proto" if c == "l7_protocol" else "int"
        tags[c] = TagDef(c, Q.SRC_U8, i, hydrate=hyd)
    for side in (0, 1):
        for j, kname in enumerate(S.KG_COLS):
            nm = f"{kname}_{side}"
            tags[nm] = TagDef(nm, Q.SRC_KG, side * S.N_KG + j)
    tags["agent_id"] = tags["vtap_id"]
    # pooled v6 addresses: L4 str cols are pooled in schema order
    for sname in ("ip6_0", "ip6_1"):
        tags[sname] = TagDef(sname, Q.SRC_STR_HASH,
                             L4.STR_COLS.index(sname), hydrate="ip6str")
    return tags


L4_TAGS = build_l4_tags()

L4_METRICS: Dict[str, TagDef] = {
    name: L4_TAGS[name] for name in [
        "byte_tx", "byte_rx", "packet_tx", "packet_rx", "total_byte_tx",
        "total_byte_rx", "rtt", "srt_sum", "srt_count", "srt_max",
        "art_sum", "art_count", "art_max", "retrans_tx", "retrans_rx",
        "l7_request", "l7_response", "l7_rrt_sum", "l7_rrt_count",
        "duration",
    ]
}
L4_METRICS["log_count"] = TagDef("log_count", Q.SRC_CONST0, 0)
