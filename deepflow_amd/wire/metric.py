"""metric protobuf schemas (wire-compatible with reference message/metric.proto).

Field numbers mirror /root/reference/message/metric.proto:14-236 (Document /
MiniTag / Meter family + Profile). Used for the agent->ingester metrics
(Document) stream and profile pipeline.
"""

MINI_FIELD = {
    1: ("ip", 'b'),
    2: ("ip1", 'b'),
    3: ("global_thread_id", 'u'),
    4: ("is_ipv6", 'u'),
    5: ("l3_epc_id", 'i'),
    6: ("l3_epc_id1", 'i'),
    7: ("mac", 'u'),
    8: ("mac1", 'u'),
    9: ("direction", 'u'),
    10: ("tap_side", 'u'),
    11: ("protocol", 'u'),
    12: ("acl_gid", 'u'),
    13: ("server_port", 'u'),
    14: ("vtap_id", 'u'),
    15: ("tap_port", 'u'),
    16: ("tap_type", 'u'),
    17: ("l7_protocol", 'u'),
    20: ("gpid", 'u'),
    21: ("gpid1", 'u'),
    22: ("signal_source", 'u'),
    23: ("app_service", 's'),
    24: ("app_instance", 's'),
    25: ("endpoint", 's'),
    27: ("pod_id", 'u'),
    28: ("biz_type", 'u'),
}

MINI_TAG = {
    1: ("field", 'm', MINI_FIELD),
    2: ("code", 'u'),
}

TRAFFIC = {
    1: ("packet_tx", 'u'), 2: ("packet_rx", 'u'), 3: ("byte_tx", 'u'), 4: ("byte_rx", 'u'),
    5: ("l3_byte_tx", 'u'), 6: ("l3_byte_rx", 'u'), 7: ("l4_byte_tx", 'u'), 8: ("l4_byte_rx", 'u'),
    9: ("new_flow", 'u'), 10: ("closed_flow", 'u'), 11: ("l7_request", 'u'),
    12: ("l7_response", 'u'), 13: ("syn", 'u'), 14: ("synack", 'u'), 15: ("direction_score", 'u'),
}

LATENCY = {
    1: ("rtt_max", 'u'), 2: ("rtt_client_max", 'u'), 3: ("rtt_server_max", 'u'),
    4: ("srt_max", 'u'), 5: ("art_max", 'u'), 6: ("rrt_max", 'u'), 19: ("cit_max", 'u'),
    7: ("rtt_sum", 'u'), 8: ("rtt_client_sum", 'u'), 9: ("rtt_server_sum", 'u'),
    10: ("srt_sum", 'u'), 11: ("art_sum", 'u'), 12: ("rrt_sum", 'u'), 20: ("cit_sum", 'u'),
    13: ("rtt_count", 'u'), 14: ("rtt_client_count", 'u'), 15: ("rtt_server_count", 'u'),
    16: ("srt_count", 'u'), 17: ("art_count", 'u'), 18: ("rrt_count", 'u'), 21: ("cit_count", 'u'),
}

PERFORMANCE = {
    1: ("retrans_tx", 'u'), 2: ("retrans_rx", 'u'), 3: ("zero_win_tx", 'u'),
    4: ("zero_win_rx", 'u'), 5: ("retrans_syn", 'u'), 6: ("retrans_synack", 'u'),
}

ANOMALY = {
    1: ("client_rst_flow", 'u'), 2: ("server_rst_flow", 'u'), 3: ("server_syn_miss", 'u'),
    4: ("client_ack_miss", 'u'), 5: ("client_half_close_flow", 'u'),
    6: ("server_half_close_flow", 'u'), 7: ("client_source_port_reuse", 'u'),
    8: ("client_establish_reset", 'u'), 9: ("server_reset", 'u'), 10: ("server_queue_lack", 'u'),
    11: ("server_establish_reset", 'u'), 12: ("tcp_timeout", 'u'),
    13: ("l7_client_error", 'u'), 14: ("l7_server_error", 'u'), 15: ("l7_timeout", 'u'),
    20: ("client_ooo", 'u'), 21: ("server_ooo", 'u'),
}

FLOW_LOAD = {1: ("load", 'u')}

FLOW_METER = {
    1: ("traffic", 'm', TRAFFIC),
    2: ("latency", 'm', LATENCY),
    3: ("performance", 'm', PERFORMANCE),
    4: ("anomaly", 'm', ANOMALY),
    5: ("flow_load", 'm', FLOW_LOAD),
}

USAGE_METER = {
    1: ("packet_tx", 'u'), 2: ("packet_rx", 'u'), 3: ("byte_tx", 'u'), 4: ("byte_rx", 'u'),
    5: ("l3_byte_tx", 'u'), 6: ("l3_byte_rx", 'u'), 7: ("l4_byte_tx", 'u'), 8: ("l4_byte_rx", 'u'),
}

APP_TRAFFIC = {1: ("request", 'u'), 2: ("response", 'u'), 3: ("direction_score", 'u')}
APP_LATENCY = {1: ("rrt_max", 'u'), 2: ("rrt_sum", 'u'), 3: ("rrt_count", 'u')}
APP_ANOMALY = {1: ("client_error", 'u'), 2: ("server_error", 'u'), 3: ("timeout", 'u')}

APP_METER = {
    1: ("traffic", 'm', APP_TRAFFIC),
    2: ("latency", 'm', APP_LATENCY),
    3: ("anomaly", 'm', APP_ANOMALY),
}

METER = {
    1: ("meter_id", 'u'),
    2: ("flow", 'm', FLOW_METER),
    3: ("usage", 'm', USAGE_METER),
    4: ("app", 'm', APP_METER),
}

DOCUMENT = {
    1: ("timestamp", 'u'),
    2: ("tag", 'm', MINI_TAG),
    3: ("meter", 'm', METER),
    4: ("flags", 'u'),
}

# ProfileEventType enum values (metric.proto:197-205)
PROFILE_EVENT_EXTERNAL = 0
PROFILE_EVENT_EBPF_ON_CPU = 1
PROFILE_EVENT_EBPF_OFF_CPU = 2
PROFILE_EVENT_EBPF_MEM_ALLOC = 3
PROFILE_EVENT_EBPF_MEM_IN_USE = 4
PROFILE_EVENT_EBPF_HBM_ALLOC = 5
PROFILE_EVENT_EBPF_HBM_IN_USE = 6
# Our roctracer-based GPU kernel profiler (net-new vs reference) reuses the
# Hbm event slots for OnGPU samples, landing in the same profile schema.

PROFILE = {
    1: ("ip", 'b'),
    2: ("name", 's'),
    3: ("units", 's'),
    4: ("aggregation_type", 's'),
    5: ("sample_rate", 'u'),
    6: ("from_time", 'u'),
    7: ("until", 'u'),
    8: ("spy_name", 's'),
    9: ("format", 's'),
    10: ("content_type", 'b'),
    11: ("data", 'b'),
    12: ("data_compressed", 'u'),
    20: ("timestamp", 'u'),
    21: ("event_type", 'u'),
    22: ("stime", 'u'),
    23: ("pid", 'u'),
    24: ("tid", 'u'),
    25: ("thread_name", 's'),
    26: ("process_name", 's'),
    27: ("u_stack_id", 'u'),
    28: ("k_stack_id", 'u'),
    29: ("cpu", 'u'),
    30: ("count", 'u'),
    33: ("pod_id", 'u'),
    34: ("wide_count", 'u'),
}

# stats.proto (dfstatsd self-metrics envelope, reference message/stats.proto:14-23)
STATS = {
    1: ("timestamp", 'u'),
    2: ("name", 's'),
    3: ("tag_names", '*s'),
    4: ("tag_values", '*s'),
    7: ("metrics_float_names", '*s'),
    8: ("metrics_float_values", '*d'),
    9: ("org_id", 'u'),
    10: ("team_id", 'u'),
}
