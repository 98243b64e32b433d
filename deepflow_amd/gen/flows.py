"""Synthetic L4 flow (TaggedFlow) generator — fixture + bench driver.

Mirrors the record shape the reference agent's FlowAggr emits
(wire schema message/flow_log.proto:14-120). Deterministic; C++ twin in
ops/csrc/gen_cpu.cpp must stay byte-identical.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List

from ..wire import pb, flow_log, framing
from .rng import SplitMix64


@dataclass
class FlowGenConfig:
    n: int = 1000
    seed: int = 2
    base_time_ns: int = 1_700_000_000_000_000_000
    dt_ns: int = 1_000_000
    n_agents: int = 8
    n_ips: int = 4096
    n_epcs: int = 16
    acl_rate_pct: int = 0   # % of flows carrying a matched ACL gid
    n_acls: int = 4
    ip6_rate_pct: int = 0   # % of flows with IPv6 addresses


def gen_flow_dict(cfg: FlowGenConfig, i: int) -> Dict:
    rng = SplitMix64(cfg.seed * 0x85EBCA6B + i)
    r0 = rng.next()
    start = cfg.base_time_ns + i * cfg.dt_ns
    dur = 1000 * (100 + rng.below(500_000))
    ip_c = 0x0A000000 | rng.below(cfg.n_ips)
    ip_s = 0x0A000000 | rng.below(cfg.n_ips)
    pkts_tx = 1 + rng.below(1000)
    pkts_rx = 1 + rng.below(1000)
    bytes_tx = pkts_tx * (64 + rng.below(1400))
    bytes_rx = pkts_rx * (64 + rng.below(1400))
    # non-rng decisions: stream stays stable when rates are 0
    v6 = cfg.ip6_rate_pct > 0 and \
        ((i * 2654435761 + cfg.seed) % 100) < cfg.ip6_rate_pct
    acl = cfg.acl_rate_pct > 0 and \
        ((i * 40503 + cfg.seed) % 100) < cfg.acl_rate_pct
    flow = {
        "flow_key": {
            "vtap_id": 1 + (r0 % cfg.n_agents),
            "tap_type": 3,
            "mac_src": r0 & 0xFFFFFFFFFFFF,
            "mac_dst": (r0 >> 8) & 0xFFFFFFFFFFFF,
            "ip_src": 0 if v6 else ip_c,
            "ip_dst": 0 if v6 else ip_s,
            "port_src": 32768 + (r0 % 28000),
            "port_dst": 443,
            "proto": 6,
        },
        "metrics_peer_src": {
            "byte_count": bytes_tx,
            "l3_byte_count": bytes_tx - 14 * pkts_tx,
            "l4_byte_count": bytes_tx - 54 * pkts_tx,
            "packet_count": pkts_tx,
            "total_byte_count": bytes_tx,
            "total_packet_count": pkts_tx,
            "first": start,
            "last": start + dur,
            "tcp_flags": 0x1B,
            "l3_epc_id": 1 + (ip_c % cfg.n_epcs),
            "is_l2_end": 1,
            "is_l3_end": 1,
        },
        "metrics_peer_dst": {
            "byte_count": bytes_rx,
            "l3_byte_count": bytes_rx - 14 * pkts_rx,
            "l4_byte_count": bytes_rx - 54 * pkts_rx,
            "packet_count": pkts_rx,
            "total_byte_count": bytes_rx,
            "total_packet_count": pkts_rx,
            "first": start,
            "last": start + dur,
            "tcp_flags": 0x1B,
            "l3_epc_id": 1 + (ip_s % cfg.n_epcs),
        },
        "flow_id": r0 & 0x7FFFFFFFFFFFFFFF,
        "start_time": start,
        "end_time": start + dur,
        "duration": dur,
        "eth_type": 0x0800,
        "has_perf_stats": 1,
        "perf_stats": {
            "tcp": {
                "rtt": 100 + rng.below(50_000),
                "srt_max": 50 + rng.below(10_000),
                "srt_sum": 50 + rng.below(100_000),
                "srt_count": 1 + rng.below(16),
                "art_max": 20 + rng.below(5_000),
                "art_sum": 20 + rng.below(50_000),
                "art_count": 1 + rng.below(16),
                "syn_count": 1,
                "synack_count": 1,
                "counts_peer_tx": {"retrans_count": rng.below(4)},
                "counts_peer_rx": {"retrans_count": rng.below(4)},
            },
            "l4_protocol": 1,
        },
        "close_type": 1,
        "signal_source": 0,
        "is_active_service": 1,
        "tap_side": 1,
        "direction_score": 255,
    }
    if v6:
        flow["flow_key"]["ip6_src"] = (b"\x20\x01\x0d\xb8" + b"\x00" * 8 +
                                       ip_c.to_bytes(4, "big"))
        flow["flow_key"]["ip6_dst"] = (b"\x20\x01\x0d\xb8" + b"\x00" * 8 +
                                       ip_s.to_bytes(4, "big"))
    if acl:
        flow["acl_gids"] = [1 + (i % cfg.n_acls)]
    return {"flow": flow}


def gen_flow_records(cfg: FlowGenConfig) -> List[bytes]:
    return [pb.encode(gen_flow_dict(cfg, i), flow_log.TAGGED_FLOW)
            for i in range(cfg.n)]


def gen_flow_payload(cfg: FlowGenConfig) -> bytes:
    return framing.pack_records(gen_flow_records(cfg))
