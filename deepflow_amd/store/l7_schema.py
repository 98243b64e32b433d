"""Python mirror of ops/csrc/l7_layout.h — column indices for the l7_flow_log
columnar layout. tests/test_layout_sync.py parses the header and asserts
these stay in sync."""

# Storage layout generation (ckissu analog): bump when any column block's
# shape or packing changes; segments record it so mixed-layout windows are
# rejected instead of misread (migration = drain + reingest, matching the
# reference's at-most-once durability posture).
LAYOUT_VERSION = 5  # v5 = binary trace/span id columns;
                    # v4 = KG joined at query time (no per-row kg block);
                    # v3 = pooled ip6 columns; v2 = packed attr pool

U64_COLS = [
    "start_time", "end_time", "flow_id", "rrt", "syscall_trace_id_request",
    "syscall_trace_id_response", "trace_id_hi", "trace_id_lo", "span_id_b",
]

U32_COLS = [
    "vtap_id", "ip4_0", "ip4_1", "l3_epc_id_0", "l3_epc_id_1", "client_port",
    "server_port", "response_code", "request_length", "response_length",
    "request_id", "process_id_0", "process_id_1", "gprocess_id_0",
    "gprocess_id_1", "pod_id_0", "pod_id_1", "req_tcp_seq", "resp_tcp_seq",
    "captured_request_byte", "captured_response_byte", "flags", "biz_type",
]

U8_COLS = [
    "tap_side", "tap_type", "protocol", "l7_protocol", "msg_type",
    "response_status", "direction_score", "is_ipv6",
]

STR_COLS = [
    "request_type", "request_domain", "request_resource", "endpoint",
    "exception_desc", "response_result", "version", "trace_id", "span_id",
    "parent_span_id", "x_request_id_0", "x_request_id_1", "http_user_agent",
    "http_referer", "service_name", "process_kname_0", "process_kname_1",
    "biz_code", "ip6_0", "ip6_1",
]

MAX_ATTRS = 16

# dict-encoded columns: (did column name, source str col index, dict domain)
DICT_DOMAINS = [
    "req_type", "domain", "resource", "endpoint", "version", "service_name",
    "attr_name", "attr_value", "exception", "pkname",
]

DID_COLS = [
    ("request_type", STR_COLS.index("request_type"), 0),
    ("request_domain", STR_COLS.index("request_domain"), 1),
    ("request_resource", STR_COLS.index("request_resource"), 2),
    ("endpoint", STR_COLS.index("endpoint"), 3),
    ("version", STR_COLS.index("version"), 4),
    ("service_name", STR_COLS.index("service_name"), 5),
]
DICT_DOM_ATTR_NAME = 6
DICT_DOM_ATTR_VALUE = 7

# KnowledgeGraph columns per side (K2 join output)
KG_COLS = [
    "pod_id", "pod_node_id", "pod_ns_id", "pod_group_id", "pod_cluster_id",
    "l3_device_type", "l3_device_id", "subnet_id", "host_id", "az_id",
    "service_id", "gprocess_id",
]

# string columns that get pooled into the segment blob (everything not
# dict-encoded); order defines K4's gather order
POOL_COLS = [
    STR_COLS.index(c) for c in [
        "exception_desc", "response_result", "trace_id", "span_id",
        "parent_span_id", "x_request_id_0", "x_request_id_1",
        "http_user_agent", "http_referer", "process_kname_0",
        "process_kname_1", "biz_code", "ip6_0", "ip6_1",
    ]
]

N_POOL = len(POOL_COLS)
# str col name -> pool position (for SRC_STR_HASH filters + select fetch)
POOL_POS = {STR_COLS[sc]: i for i, sc in enumerate(POOL_COLS)}

DICT_ID_INVALID = 0xFFFFFFFF

N_U64 = len(U64_COLS)
N_U32 = len(U32_COLS)
N_U8 = len(U8_COLS)
N_STR = len(STR_COLS)
N_DID = len(DID_COLS)
N_KG = len(KG_COLS)


def str_ref_pack(off: int, ln: int) -> int:
    return (off << 16) | min(ln, 0xFFFF)


def str_ref_unpack(r: int):
    return r >> 16, r & 0xFFFF
