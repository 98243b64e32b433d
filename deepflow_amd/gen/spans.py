"""Synthetic L7 span (AppProtoLogsData) stream generator.

Test fixture + benchmark driver for the ingest pipeline (BASELINE config #2:
10M-span synthetic stream with a 100k-cardinality tag dictionary). Mirrors the
record shapes the reference agent's AppProtoLogsParser emits
(agent/src/flow_generator/protocol_logs/pb_adapter.rs behavior; wire schema
message/flow_log.proto:224-311).

Determinism contract: given the same SpanGenConfig, the C++ generator
(ops/csrc/gen_cpu.cpp) must produce byte-identical payloads; tests/test_gen.py
golden-checks this. Keep any change to the field logic mirrored there.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List

from ..wire import pb, flow_log, framing
from ..wire.const_enums import (
    L7_PROTOCOL_HTTP_1, STATUS_OK, STATUS_SERVER_ERROR, SIGNAL_SOURCE_EBPF,
)
from .rng import SplitMix64


@dataclass
class SpanGenConfig:
    n: int = 1000
    seed: int = 1
    base_time_ns: int = 1_700_000_000_000_000_000
    dt_ns: int = 1_000_000          # spacing between span start times
    n_agents: int = 8
    n_ips: int = 4096               # distinct pod/service IPs
    n_epcs: int = 16
    n_services: int = 256           # domain/endpoint cardinality
    n_resources: int = 4096         # URL path cardinality
    tag_cardinality: int = 100_000  # custom attribute value cardinality
    n_attrs: int = 4                # custom attributes per span
    err_rate_pct: int = 2
    ip6_rate_pct: int = 0           # % of spans carrying IPv6 addresses


HEX = "0123456789abcdef"


def _hex(v: int, width: int) -> str:
    out = []
    for i in range(width):
        out.append(HEX[(v >> (4 * (width - 1 - i))) & 0xF])
    return "".join(out)


def gen_span_dict(cfg: SpanGenConfig, i: int) -> Dict:
    """Generate span #i deterministically (one rng stream per record)."""
    rng = SplitMix64(cfg.seed * 0x9E3779B9 + i)
    r0 = rng.next()
    start = cfg.base_time_ns + i * cfg.dt_ns + rng.below(1000) * 1000
    rrt_us = 100 + rng.below(200_000)         # 0.1ms .. 200ms
    end = start + rrt_us * 1000
    svc = rng.below(cfg.n_services)
    res = rng.below(cfg.n_resources)
    ip_c = 0x0A000000 | rng.below(cfg.n_ips)
    ip_s = 0x0A000000 | (svc * 7 % cfg.n_ips)
    # v6 decision draws nothing from the record rng stream so the byte
    # stream with ip6_rate_pct=0 is unchanged (golden-test stability)
    v6 = cfg.ip6_rate_pct > 0 and \
        ((i * 2654435761 + cfg.seed) % 100) < cfg.ip6_rate_pct
    err = rng.below(100) < cfg.err_rate_pct
    trace_hi, trace_lo = rng.next(), rng.next()
    span_id_v = rng.next()

    base = {
        "start_time": start,
        "end_time": end,
        "flow_id": r0 & 0x7FFFFFFFFFFFFFFF,
        "tap_port": 0,
        "vtap_id": 1 + (r0 % cfg.n_agents),
        "tap_type": 3,
        "tap_side": 1,  # client-side
        "head": {"proto": L7_PROTOCOL_HTTP_1, "msg_type": 2, "rrt": rrt_us},
        "ip_src": 0 if v6 else ip_c,
        "ip_dst": 0 if v6 else ip_s,
        "l3_epc_id_src": 1 + (ip_c % cfg.n_epcs),
        "l3_epc_id_dst": 1 + (ip_s % cfg.n_epcs),
        "port_src": 32768 + (r0 % 28000),
        "port_dst": 8080,
        "protocol": 6,
        "req_tcp_seq": r0 & 0xFFFFFFFF,
        "resp_tcp_seq": (r0 >> 16) & 0xFFFFFFFF,
        "process_id_0": 1000 + rng.below(64),
        "process_id_1": 2000 + (svc % 64),
        "syscall_trace_id_request": rng.next() & 0x7FFFFFFFFFFFFFFF,
        "gpid_0": 1 + rng.below(1 << 16),
        "gpid_1": 1 + (svc % (1 << 16)),
        "pod_id_0": 1 + (ip_c % cfg.n_ips),
        "pod_id_1": 1 + (ip_s % cfg.n_ips),
    }
    if v6:
        base["is_ipv6"] = 1
        base["ip6_src"] = (b"\x20\x01\x0d\xb8" + b"\x00" * 8 +
                           ip_c.to_bytes(4, "big"))
        base["ip6_dst"] = (b"\x20\x01\x0d\xb8" + b"\x00" * 8 +
                           ip_s.to_bytes(4, "big"))
    attrs_n: List[str] = []
    attrs_v: List[str] = []
    for a in range(cfg.n_attrs):
        attrs_n.append("attr_%d" % a)
        attrs_v.append("v%07d" % rng.below(cfg.tag_cardinality))
    span = {
        "base": base,
        "req_len": 128 + rng.below(1024),
        "resp_len": 256 + rng.below(8192),
        "req": {
            "req_type": "POST" if (r0 >> 8) % 4 == 0 else "GET",
            "domain": "svc-%03d.example.com" % svc,
            "resource": "/api/v1/r/%05d" % res,
            "endpoint": "/api/v1/r",
        },
        "resp": {
            "status": STATUS_SERVER_ERROR if err else STATUS_OK,
            "code": 500 if err else 200,
        },
        "version": "1.1",
        "trace_info": {
            "trace_id": _hex(trace_hi, 16) + _hex(trace_lo, 16),
            "span_id": _hex(span_id_v, 16),
        },
        "ext_info": {
            "service_name": "svc-%03d" % svc,
            "request_id": rng.below(1 << 30),
            "attribute_names": attrs_n,
            "attribute_values": attrs_v,
        },
        "direction_score": 255,
        "captured_request_byte": 128,
        "captured_response_byte": 256,
    }
    return span


def gen_span_records(cfg: SpanGenConfig) -> List[bytes]:
    return [pb.encode(gen_span_dict(cfg, i), flow_log.APP_PROTO_LOGS_DATA)
            for i in range(cfg.n)]


def gen_span_payload(cfg: SpanGenConfig) -> bytes:
    """Length-prefixed record payload (the MSG_PROTOCOLLOG frame body)."""
    return framing.pack_records(gen_span_records(cfg))
