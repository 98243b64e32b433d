// Column layout ABI for decoded L4 flow logs (TaggedFlow).
// Mirrors the reference l4_flow_log column groups
// (server/ingester/flow_log/log_data/l4_flow_log.go:467-455): DataLink,
// Network, Transport (incl. TCP perf), Application (L7 perf), FlowInfo,
// Metrics. Mirrored in deepflow_amd/store/l4_schema.py (sync test).
#pragma once
#include <stdint.h>

enum {
    L4_U64_START_TIME = 0,
    L4_U64_END_TIME,
    L4_U64_DURATION,
    L4_U64_FLOW_ID,
    L4_U64_MAC_SRC,
    L4_U64_MAC_DST,
    L4_U64_BYTE_TX,
    L4_U64_BYTE_RX,
    L4_U64_L3_BYTE_TX,
    L4_U64_L3_BYTE_RX,
    L4_U64_L4_BYTE_TX,
    L4_U64_L4_BYTE_RX,
    L4_U64_PACKET_TX,
    L4_U64_PACKET_RX,
    L4_U64_TOTAL_BYTE_TX,
    L4_U64_TOTAL_BYTE_RX,
    L4_U64_TOTAL_PACKET_TX,
    L4_U64_TOTAL_PACKET_RX,
    L4_U64_L7_RRT_SUM,
    L4_U64_N
};

enum {
    L4_U32_VTAP_ID = 0,
    L4_U32_IP4_0,
    L4_U32_IP4_1,
    L4_U32_EPC_0,
    L4_U32_EPC_1,
    L4_U32_PORT_SRC,
    L4_U32_PORT_DST,
    L4_U32_TCP_FLAGS_SRC,
    L4_U32_TCP_FLAGS_DST,
    L4_U32_RTT,
    L4_U32_SRT_SUM,
    L4_U32_SRT_COUNT,
    L4_U32_SRT_MAX,
    L4_U32_ART_SUM,
    L4_U32_ART_COUNT,
    L4_U32_ART_MAX,
    L4_U32_CIT_SUM,
    L4_U32_CIT_COUNT,
    L4_U32_CIT_MAX,
    L4_U32_RETRANS_TX,
    L4_U32_RETRANS_RX,
    L4_U32_ZERO_WIN_TX,
    L4_U32_ZERO_WIN_RX,
    L4_U32_OOO_TX,
    L4_U32_OOO_RX,
    L4_U32_SYN_COUNT,
    L4_U32_SYNACK_COUNT,
    L4_U32_RETRANS_TOTAL,
    L4_U32_L7_REQUEST,
    L4_U32_L7_RESPONSE,
    L4_U32_L7_RRT_COUNT,
    L4_U32_L7_RRT_MAX,
    L4_U32_L7_ERR_CLIENT,
    L4_U32_L7_ERR_SERVER,
    L4_U32_L7_ERR_TIMEOUT,
    L4_U32_GPID_0,
    L4_U32_GPID_1,
    L4_U32_NAT_REAL_IP_0,
    L4_U32_NAT_REAL_IP_1,
    L4_U32_NAT_REAL_PORT_0,
    L4_U32_NAT_REAL_PORT_1,
    L4_U32_VLAN,
    L4_U32_ETH_TYPE,
    L4_U32_ACL_GID,      // first matched ACL group id (traffic_policy key)
    L4_U32_N
};

enum {
    L4_U8_CLOSE_TYPE = 0,
    L4_U8_TAP_SIDE,
    L4_U8_TAP_TYPE,
    L4_U8_PROTOCOL,
    L4_U8_L4_PROTOCOL,
    L4_U8_L7_PROTOCOL,
    L4_U8_SIGNAL_SOURCE,
    L4_U8_IS_NEW_FLOW,
    L4_U8_IS_ACTIVE_SERVICE,
    L4_U8_DIRECTION_SCORE,
    L4_U8_N
};

enum {
    L4_STR_REQUEST_DOMAIN = 0,
    L4_STR_IP6_0,        // 16-byte v6 address, pooled (empty for v4 rows)
    L4_STR_IP6_1,
    L4_STR_N
};
