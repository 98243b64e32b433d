"""application.1s metric rollup store (K5 output).

GPU mode: open-addressing key/accumulator tensors written by k_agg_app1s.
CPU mode: plain dict (ops/ref.agg_app1s_ref). Key packing (must match
dfgpu.hip k_agg_app1s):
  key = rel_s<<42 | (vtap&0xFFF)<<30 | l7proto<<22 | (status&0xF)<<18
        | (server_port&0xFFFF)<<2 | 1
"""
from __future__ import annotations

from typing import Dict, List

import torch

AGG_FIELDS = ["request", "response", "client_error", "server_error",
              "rrt_sum", "rrt_count", "rrt_max"]
AGG_NVALS = len(AGG_FIELDS)


def unpack_key(key: int, time_base_s: int) -> Dict[str, int]:
    return {
        "time": time_base_s + (key >> 42),
        "vtap_id": (key >> 30) & 0xFFF,
        "l7_protocol": (key >> 22) & 0xFF,
        "response_status": (key >> 18) & 0xF,
        "server_port": (key >> 2) & 0xFFFF,
    }


NET_FIELDS = ["byte_tx", "byte_rx", "packet_tx", "packet_rx", "new_flow",
              "closed_flow", "rtt_sum", "rtt_count", "rtt_max", "retrans"]


def unpack_net_key(key: int, time_base_s: int):
    return {
        "time": time_base_s + (key >> 40),
        "vtap_id": (key >> 28) & 0xFFF,
        "l3_epc_id": (key >> 12) & 0xFFFF,
        "protocol": (key >> 4) & 0xFF,
    }




class _MetricsCkpt:
    """Checkpoint mixin shared by the 1s rollup tables."""

    def state_dict(self):
        if self.device == "cpu":
            return {"table": {k: list(v) for k, v in self.table.items()}}
        return {"tkeys": self.tkeys.cpu().clone(),
                "tvals": self.tvals.cpu().clone()}

    def load_state_dict(self, st):
        if self.device == "cpu":
            self.table.update(st["table"])
        else:
            self.tkeys.copy_(st["tkeys"].to(self.tkeys.device))
            self.tvals.copy_(st["tvals"].to(self.tvals.device))


class Net1sMetrics(_MetricsCkpt):
    """network.1s rollup (K5b output; reference flow_metrics network table)."""

    def __init__(self, time_base_s: int, capacity_pow2: int = 1 << 20,
                 device: str = "cpu"):
        assert capacity_pow2 & (capacity_pow2 - 1) == 0
        self.time_base_s = time_base_s
        self.device = device
        self.capacity = capacity_pow2
        if device == "cpu":
            self.table: Dict[int, List[int]] = {}
        else:
            dev = torch.device(device)
            self.tkeys = torch.zeros(capacity_pow2, dtype=torch.int64, device=dev)
            self.tvals = torch.zeros((capacity_pow2, len(NET_FIELDS)),
                                     dtype=torch.int64, device=dev)

    def rows(self) -> List[Dict[str, int]]:
        out = []
        if self.device == "cpu":
            for key, acc in self.table.items():
                row = unpack_net_key(key, self.time_base_s)
                row.update(dict(zip(NET_FIELDS, acc)))
                out.append(row)
        else:
            mask = self.tkeys != 0
            keys = self.tkeys[mask].cpu().numpy()
            vals = self.tvals[mask].cpu().numpy()
            for key, acc in zip(keys, vals):
                row = unpack_net_key(int(key) & ((1 << 64) - 1),
                                     self.time_base_s)
                row.update({f: int(a) for f, a in zip(NET_FIELDS, acc)})
                out.append(row)
        out.sort(key=lambda r: (r["time"], r["vtap_id"], r["l3_epc_id"],
                                r["protocol"]))
        return out


class App1sMetrics(_MetricsCkpt):
    def __init__(self, time_base_s: int, capacity_pow2: int = 1 << 20,
                 device: str = "cpu"):
        assert capacity_pow2 & (capacity_pow2 - 1) == 0
        self.time_base_s = time_base_s
        self.device = device
        self.capacity = capacity_pow2
        if device == "cpu":
            self.table: Dict[int, List[int]] = {}
        else:
            dev = torch.device(device)
            self.tkeys = torch.zeros(capacity_pow2, dtype=torch.int64, device=dev)
            self.tvals = torch.zeros((capacity_pow2, AGG_NVALS),
                                     dtype=torch.int64, device=dev)

    def rows(self) -> List[Dict[str, int]]:
        out = []
        if self.device == "cpu":
            items = self.table.items()
            for key, acc in items:
                row = unpack_key(key, self.time_base_s)
                row.update(dict(zip(AGG_FIELDS, acc)))
                out.append(row)
        else:
            mask = self.tkeys != 0
            keys = self.tkeys[mask].cpu().numpy()
            vals = self.tvals[mask].cpu().numpy()
            for key, acc in zip(keys, vals):
                row = unpack_key(int(key) & ((1 << 64) - 1), self.time_base_s)
                row.update({f: int(a) for f, a in zip(AGG_FIELDS, acc)})
                out.append(row)
        out.sort(key=lambda r: (r["time"], r["vtap_id"], r["server_port"],
                                r["l7_protocol"], r["response_status"]))
        return out
