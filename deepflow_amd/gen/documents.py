"""Synthetic Document (flow_metrics) generator — fixture + bench driver.

Mirrors the agent collector's Document output (wire schema
message/metric.proto:51-68; agent/src/metric/document.rs behavior).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List

from ..wire import pb, metric, framing
from .rng import SplitMix64


@dataclass
class DocGenConfig:
    n: int = 1000
    seed: int = 3
    base_time_s: int = 1_700_000_000
    n_agents: int = 8
    n_ips: int = 4096
    n_epcs: int = 16


def gen_document_dict(cfg: DocGenConfig, i: int) -> Dict:
    rng = SplitMix64(cfg.seed * 0xC2B2AE35 + i)
    r0 = rng.next()
    ip = 0x0A000000 | rng.below(cfg.n_ips)
    req = 1 + rng.below(100)
    err = rng.below(8)
    rrt_max = 100 + rng.below(100_000)
    doc = {
        "timestamp": cfg.base_time_s + (i % 60),
        "tag": {
            "field": {
                "ip": bytes([10, (ip >> 16) & 0xFF, (ip >> 8) & 0xFF, ip & 0xFF]),
                "l3_epc_id": 1 + (ip % cfg.n_epcs),
                "direction": 1,
                "protocol": 6,
                "server_port": 8080,
                "vtap_id": 1 + (r0 % cfg.n_agents),
                "tap_type": 3,
                "l7_protocol": 20,
                "signal_source": 3,
            },
            # code bitmask: which tag fields are populated (reference
            # libs/flow-metrics Code semantics); IP|L3EpcID|Direction|
            # Protocol|ServerPort|VTAPID|TAPType composition for the
            # application.1s table.
            "code": 0x3F,
        },
        "meter": {
            "meter_id": 4,  # app meter
            "app": {
                "traffic": {"request": req, "response": req, "direction_score": 255},
                "latency": {
                    "rrt_max": rrt_max,
                    "rrt_sum": req * (rrt_max // 2),
                    "rrt_count": req,
                },
                "anomaly": {"server_error": err, "timeout": rng.below(2)},
            },
        },
        "flags": 0,
    }
    return doc


def gen_document_records(cfg: DocGenConfig) -> List[bytes]:
    return [pb.encode(gen_document_dict(cfg, i), metric.DOCUMENT)
            for i in range(cfg.n)]


def gen_document_payload(cfg: DocGenConfig) -> bytes:
    return framing.pack_records(gen_document_records(cfg))
