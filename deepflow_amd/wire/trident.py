"""trident.proto schema subset (wire-compatible field numbers).

Mirrors /root/reference/message/trident.proto for the Synchronizer RPC
surface this controller implements: SyncRequest (:85-125), SyncResponse
(:594-622), Config (:209-257 subset), PlatformData/Interface/Cidr
(:385-503 subset), VtapIp (:569-575). Carried over gRPC (HTTP/2) by
control/grpc_server.py.
"""

SYNC_REQUEST = {
    1: ("boot_time", 'u'),
    2: ("config_accepted", 'u'),
    5: ("revision", 's'),
    6: ("exception", 'u'),
    7: ("process_name", 's'),
    9: ("version_platform_data", 'u'),
    10: ("version_acls", 'u'),
    11: ("version_groups", 'u'),
    21: ("ctrl_ip", 's'),
    22: ("host", 's'),
    23: ("host_ips", '*s'),
    25: ("ctrl_mac", 's'),
    26: ("vtap_group_id_request", 's'),
    29: ("team_id", 's'),
    32: ("cpu_num", 'u'),
    33: ("memory_size", 'u'),
    34: ("arch", 's'),
    35: ("os", 's'),
    36: ("kernel_version", 's'),
    50: ("org_id", 'u'),
}

CONFIG = {
    1: ("enabled", 'u'),
    2: ("max_cpus", 'u'),
    3: ("max_memory", 'u'),
    4: ("sync_interval", 'u'),
    5: ("stats_interval", 'u'),
    6: ("global_pps_threshold", 'u'),
    8: ("tap_interface_regex", 's'),
    40: ("vtap_id", 'u'),
}

IP_RESOURCE = {
    1: ("ip", 's'),
    2: ("masklen", 'u'),
    3: ("subnet_id", 'u'),
}

INTERFACE = {
    1: ("id", 'u'),
    2: ("device_type", 'u'),
    3: ("device_id", 'u'),
    4: ("if_type", 'u'),
    6: ("epc_id", 'u'),
    7: ("launch_server", 's'),
    8: ("ip_resources", '*m', IP_RESOURCE),
    9: ("launch_server_id", 'u'),
}

CIDR = {
    1: ("prefix", 's'),
    2: ("type", 'u'),
    3: ("epc_id", 'i'),
    4: ("subnet_id", 'u'),
    5: ("region_id", 'u'),
    6: ("az_id", 'u'),
}

PLATFORM_DATA = {
    1: ("interfaces", '*m', INTERFACE),
    4: ("cidrs", '*m', CIDR),
}

VTAP_IP = {
    1: ("vtap_id", 'u'),
    2: ("epc_id", 'u'),
    3: ("ip", 's'),
    5: ("team_id", 'u'),
    6: ("org_id", 'u'),
}

SYNC_RESPONSE = {
    1: ("status", 'u'),
    2: ("config", 'm', CONFIG),
    4: ("revision", 's'),
    6: ("version_platform_data", 'u'),
    7: ("version_acls", 'u'),
    8: ("version_groups", 'u'),
    12: ("platform_data", 'b'),
    13: ("flow_acls", 'b'),
    15: ("groups", 'b'),
    18: ("vtap_ips", '*m', VTAP_IP),
}

STATUS_SUCCESS = 0
STATUS_FAILED = 1
STATUS_HEARTBEAT = 2


# Upgrade stream chunk (trident.proto UpgradeResponse)
UPGRADE_RESPONSE = {
    1: ("status", 'u'),
    2: ("content", 'b'),
    3: ("md5", 's'),
    4: ("pkt_count", 'u'),
    5: ("total_len", 'u'),
}
