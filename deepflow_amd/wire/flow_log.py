"""flow_log protobuf schemas (wire-compatible with reference message/flow_log.proto).

Field numbers mirror /root/reference/message/flow_log.proto:14-311 exactly;
this file defines the ABI for agent->ingester L4 flow logs (TaggedFlow) and
L7 request logs (AppProtoLogsData). See deepflow_amd/wire/pb.py for kinds.
"""

FLOW_KEY = {
    1: ("vtap_id", 'u'),
    2: ("tap_type", 'u'),
    3: ("tap_port", 'u'),
    4: ("mac_src", 'u'),
    5: ("mac_dst", 'u'),
    6: ("ip_src", 'u'),
    7: ("ip_dst", 'u'),
    8: ("ip6_src", 'b'),
    9: ("ip6_dst", 'b'),
    10: ("port_src", 'u'),
    11: ("port_dst", 'u'),
    12: ("proto", 'u'),
}

FLOW_METRICS_PEER = {
    1: ("byte_count", 'u'),
    2: ("l3_byte_count", 'u'),
    3: ("l4_byte_count", 'u'),
    4: ("packet_count", 'u'),
    5: ("total_byte_count", 'u'),
    6: ("total_packet_count", 'u'),
    7: ("first", 'u'),
    8: ("last", 'u'),
    9: ("tcp_flags", 'u'),
    10: ("l3_epc_id", 'i'),
    11: ("is_l2_end", 'u'),
    12: ("is_l3_end", 'u'),
    13: ("is_active_host", 'u'),
    14: ("is_device", 'u'),
    15: ("is_vip_interface", 'u'),
    16: ("is_vip", 'u'),
    20: ("real_ip", 'u'),
    21: ("real_port", 'u'),
    22: ("gpid", 'u'),
}

TUNNEL_FIELD = {
    1: ("tx_ip0", 'u'), 2: ("tx_ip1", 'u'), 3: ("rx_ip0", 'u'), 4: ("rx_ip1", 'u'),
    5: ("tx_mac0", 'u'), 6: ("tx_mac1", 'u'), 7: ("rx_mac0", 'u'), 8: ("rx_mac1", 'u'),
    9: ("tx_id", 'u'), 10: ("rx_id", 'u'), 11: ("tunnel_type", 'u'), 12: ("tier", 'u'),
    13: ("is_ipv6", 'u'),
}

TCP_PERF_COUNTS_PEER = {
    1: ("retrans_count", 'u'),
    2: ("zero_win_count", 'u'),
    3: ("ooo_count", 'u'),
}

TCP_PERF_STATS = {
    1: ("rtt_client_max", 'u'),
    2: ("rtt_server_max", 'u'),
    3: ("srt_max", 'u'),
    4: ("art_max", 'u'),
    5: ("rtt", 'u'),
    8: ("srt_sum", 'u'),
    9: ("art_sum", 'u'),
    12: ("srt_count", 'u'),
    13: ("art_count", 'u'),
    14: ("counts_peer_tx", 'm', TCP_PERF_COUNTS_PEER),
    15: ("counts_peer_rx", 'm', TCP_PERF_COUNTS_PEER),
    16: ("total_retrans_count", 'u'),
    17: ("syn_count", 'u'),
    18: ("synack_count", 'u'),
    19: ("cit_max", 'u'),
    20: ("cit_sum", 'u'),
    21: ("cit_count", 'u'),
    22: ("fin_count", 'u'),
}

L7_PERF_STATS = {
    1: ("request_count", 'u'),
    2: ("response_count", 'u'),
    3: ("err_client_count", 'u'),
    4: ("err_server_count", 'u'),
    5: ("err_timeout", 'u'),
    6: ("rrt_count", 'u'),
    7: ("rrt_sum", 'u'),
    8: ("rrt_max", 'u'),
    9: ("tls_rtt", 'u'),
}

FLOW_PERF_STATS = {
    1: ("tcp", 'm', TCP_PERF_STATS),
    2: ("l7", 'm', L7_PERF_STATS),
    3: ("l4_protocol", 'u'),
    4: ("l7_protocol", 'u'),
    5: ("l7_failed_count", 'u'),
}

FLOW = {
    1: ("flow_key", 'm', FLOW_KEY),
    2: ("metrics_peer_src", 'm', FLOW_METRICS_PEER),
    3: ("metrics_peer_dst", 'm', FLOW_METRICS_PEER),
    4: ("tunnel", 'm', TUNNEL_FIELD),
    5: ("flow_id", 'u'),
    6: ("start_time", 'u'),
    7: ("end_time", 'u'),
    8: ("duration", 'u'),
    10: ("vlan", 'u'),
    11: ("eth_type", 'u'),
    12: ("has_perf_stats", 'u'),
    13: ("perf_stats", 'm', FLOW_PERF_STATS),
    14: ("close_type", 'u'),
    15: ("signal_source", 'u'),
    16: ("is_active_service", 'u'),
    17: ("queue_hash", 'u'),
    18: ("is_new_flow", 'u'),
    19: ("tap_side", 'u'),
    20: ("syn_seq", 'u'),
    21: ("synack_seq", 'u'),
    22: ("last_keepalive_seq", 'u'),
    23: ("last_keepalive_ack", 'u'),
    24: ("acl_gids", '*u'),
    25: ("direction_score", 'u'),
    26: ("request_domain", 's'),
    27: ("aggregated_flow_ids", '*u'),
    28: ("init_ipid", 'u'),
}

TAGGED_FLOW = {
    1: ("flow", 'm', FLOW),
}

L7_REQUEST = {
    1: ("req_type", 's'),
    2: ("domain", 's'),
    3: ("resource", 's'),
    4: ("endpoint", 's'),
}

L7_RESPONSE = {
    1: ("status", 'u'),
    2: ("code", 'i'),
    3: ("exception", 's'),
    4: ("result", 's'),
}

TRACE_INFO = {
    1: ("trace_id", 's'),
    2: ("span_id", 's'),
    3: ("parent_span_id", 's'),
    4: ("trace_ids", '*s'),
}

EXTENDED_INFO = {
    1: ("service_name", 's'),
    2: ("client_ip", 's'),
    3: ("request_id", 'u'),
    4: ("x_request_id_0", 's'),
    6: ("http_user_agent", 's'),
    7: ("http_referer", 's'),
    8: ("rpc_service", 's'),
    9: ("protocol_str", 's'),
    10: ("x_request_id_1", 's'),
    16: ("attribute_names", '*s'),
    17: ("attribute_values", '*s'),
    18: ("metrics_names", '*s'),
    19: ("metrics_values", '*d'),
}

APP_PROTO_HEAD = {
    1: ("proto", 'u'),
    2: ("msg_type", 'u'),
    5: ("rrt", 'u'),
}

APP_PROTO_LOGS_BASE_INFO = {
    1: ("start_time", 'u'),
    2: ("end_time", 'u'),
    3: ("flow_id", 'u'),
    4: ("tap_port", 'u'),
    5: ("vtap_id", 'u'),
    6: ("tap_type", 'u'),
    7: ("is_ipv6", 'u'),
    8: ("tap_side", 'u'),
    9: ("head", 'm', APP_PROTO_HEAD),
    10: ("mac_src", 'u'),
    11: ("mac_dst", 'u'),
    12: ("ip_src", 'u'),
    13: ("ip_dst", 'u'),
    14: ("ip6_src", 'b'),
    15: ("ip6_dst", 'b'),
    16: ("l3_epc_id_src", 'i'),
    17: ("l3_epc_id_dst", 'i'),
    18: ("port_src", 'u'),
    19: ("port_dst", 'u'),
    20: ("protocol", 'u'),
    21: ("is_vip_interface_src", 'u'),
    22: ("is_vip_interface_dst", 'u'),
    23: ("req_tcp_seq", 'u'),
    24: ("resp_tcp_seq", 'u'),
    25: ("process_id_0", 'u'),
    26: ("process_id_1", 'u'),
    27: ("process_kname_0", 's'),
    28: ("process_kname_1", 's'),
    29: ("syscall_trace_id_request", 'u'),
    30: ("syscall_trace_id_response", 'u'),
    31: ("syscall_trace_id_thread_0", 'u'),
    32: ("syscall_trace_id_thread_1", 'u'),
    33: ("syscall_cap_seq_0", 'u'),
    34: ("syscall_cap_seq_1", 'u'),
    35: ("gpid_0", 'u'),
    36: ("gpid_1", 'u'),
    39: ("syscall_coroutine_0", 'u'),
    40: ("syscall_coroutine_1", 'u'),
    41: ("pod_id_0", 'u'),
    42: ("pod_id_1", 'u'),
    43: ("biz_type", 'u'),
}

APP_PROTO_LOGS_DATA = {
    1: ("base", 'm', APP_PROTO_LOGS_BASE_INFO),
    9: ("req_len", 'i'),
    10: ("resp_len", 'i'),
    11: ("req", 'm', L7_REQUEST),
    12: ("resp", 'm', L7_RESPONSE),
    13: ("version", 's'),
    14: ("trace_info", 'm', TRACE_INFO),
    15: ("ext_info", 'm', EXTENDED_INFO),
    16: ("row_effect", 'u'),
    17: ("direction_score", 'u'),
    18: ("flags", 'u'),
    19: ("captured_request_byte", 'u'),
    20: ("captured_response_byte", 'u'),
    21: ("biz_code", 's'),
    22: ("biz_scenario", 's'),
    23: ("biz_response_code", 's'),
}
