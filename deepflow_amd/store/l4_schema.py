"""Python mirror of ops/csrc/l4_layout.h (sync-tested)."""

U64_COLS = [
    "start_time", "end_time", "duration", "flow_id", "mac_src", "mac_dst",
    "byte_tx", "byte_rx", "l3_byte_tx", "l3_byte_rx", "l4_byte_tx",
    "l4_byte_rx", "packet_tx", "packet_rx", "total_byte_tx", "total_byte_rx",
    "total_packet_tx", "total_packet_rx", "l7_rrt_sum",
]

U32_COLS = [
    "vtap_id", "ip4_0", "ip4_1", "l3_epc_id_0", "l3_epc_id_1", "client_port",
    "server_port", "tcp_flags_bit_0", "tcp_flags_bit_1", "rtt", "srt_sum",
    "srt_count", "srt_max", "art_sum", "art_count", "art_max", "cit_sum",
    "cit_count", "cit_max", "retrans_tx", "retrans_rx", "zero_win_tx",
    "zero_win_rx", "ooo_tx", "ooo_rx", "syn_count", "synack_count",
    "retrans_total", "l7_request", "l7_response", "l7_rrt_count",
    "l7_rrt_max", "l7_err_client", "l7_err_server", "l7_err_timeout",
    "gprocess_id_0", "gprocess_id_1", "nat_real_ip_0", "nat_real_ip_1",
    "nat_real_port_0", "nat_real_port_1", "vlan", "eth_type", "acl_gid",
]

U8_COLS = [
    "close_type", "tap_side", "tap_type", "protocol", "l4_protocol",
    "l7_protocol", "signal_source", "is_new_flow", "is_active_service",
    "direction_score",
]

STR_COLS = ["request_domain", "ip6_0", "ip6_1"]

KG_COLS = None  # shares l7_schema.KG_COLS

N_U64 = len(U64_COLS)
N_U32 = len(U32_COLS)
N_U8 = len(U8_COLS)
N_STR = len(STR_COLS)
