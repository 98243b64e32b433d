"""KnowledgeGraph platform-data table: (l3_epc_id, ipv4) -> resource IDs.

The GPU twin of the reference ingester's PlatformInfoTable
(server/libs/grpc/grpc_platformdata.go:147-376): the controller-lite pushes
(epc, ip) -> {pod, node, namespace, group, cluster, device, subnet, host,
az, service, gprocess} rows; the tag-join kernel (K2) probes this table
twice per span (client+server side) to fill the universal-tag columns.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Tuple

import numpy as np
import torch

from . import l7_schema as S


@dataclass
class KgInfo:
    pod_id: int = 0
    pod_node_id: int = 0
    pod_ns_id: int = 0
    pod_group_id: int = 0
    pod_cluster_id: int = 0
    l3_device_type: int = 0
    l3_device_id: int = 0
    subnet_id: int = 0
    host_id: int = 0
    az_id: int = 0
    service_id: int = 0
    gprocess_id: int = 0

    def as_list(self):
        return [self.pod_id, self.pod_node_id, self.pod_ns_id,
                self.pod_group_id, self.pod_cluster_id, self.l3_device_type,
                self.l3_device_id, self.subnet_id, self.host_id, self.az_id,
                self.service_id, self.gprocess_id]


class KnowledgeGraphTable:
    def __init__(self, capacity_pow2: int = 1 << 20, device: str = "cpu"):
        assert capacity_pow2 & (capacity_pow2 - 1) == 0
        self.capacity = capacity_pow2
        self.device = device
        dev = torch.device(device)
        self.tkeys = torch.zeros(capacity_pow2, dtype=torch.int64, device=dev)
        self.tvals = torch.zeros((capacity_pow2, S.N_KG), dtype=torch.int32,
                                 device=dev)
        # host mirror for CPU reference + rebuilds
        self.host: Dict[Tuple[int, int], KgInfo] = {}
        self.version = 0

    def update(self, entries: Dict[Tuple[int, int], KgInfo]) -> None:
        """Apply a platform-data push (epc, ip) -> info; idempotent upsert."""
        self.host.update(entries)
        self.version += 1
        if not entries:
            return
        keys = np.array(
            [((epc & 0xFFFFFFFF) << 32) | (ip & 0xFFFFFFFF)
             for epc, ip in entries.keys()], dtype=np.uint64)
        vals = np.array([v.as_list() for v in entries.values()],
                        dtype=np.int32)
        kt = torch.from_numpy(keys.view(np.int64)).to(self.tkeys.device)
        vt = torch.from_numpy(vals).to(self.tvals.device)
        if self.device == "cpu":
            from ..ops import ref
            ref.kg_build_ref(kt, vt, self.tkeys, self.tvals)
        else:
            from ..ops import gpu_ops
            gpu_ops.kg_build(kt, vt, self.tkeys, self.tvals)

    def lookup(self, epc: int, ip: int) -> KgInfo:
        return self.host.get((epc, ip), KgInfo())


def kg_display_name(col: str, ident: int) -> str:
    """Synthetic tagrecorder inventory: the display name a resource id
    hydrates to (k8s-realistic shapes; deterministic). This is the string
    a naive ClickHouse schema would store PER ROW for each universal tag
    — the SmartEncoding baseline measures the bytes of exactly these
    names (reference: tagrecorder ch_* maps + dictGet hydration)."""
    h = (ident * 0x9E3779B9) & 0xFFFFFFFFFF
    if col == "pod_id":
        return f"svc-{ident % 997:03d}-{h:010x}-{ident % 99999:05d}"
    if col == "pod_node_id":
        return f"node-us-east-{ident:04d}.prod.internal"
    if col == "pod_ns_id":
        return f"ns-team-{ident:03d}"
    if col == "pod_group_id":
        return f"svc-{ident % 997:03d}-deployment"
    if col == "pod_cluster_id":
        return f"prod-cluster-{ident:02d}"
    if col == "l3_device_type":
        return "pod" if ident == 14 else f"devtype-{ident}"
    if col == "l3_device_id":
        return f"vm-{h:010x}"
    if col == "subnet_id":
        return f"subnet-10-{ident:03d}-0-0"
    if col == "host_id":
        return f"host-{ident:04d}.dc1.example.com"
    if col == "az_id":
        return f"az-east-{ident}"
    if col == "service_id":
        return f"svc-{ident % 997:03d}.default.svc"
    if col == "gprocess_id":
        return f"proc-{h:08x}-{ident:05d}"
    return f"{col}-{ident}"


def default_platform(cfg) -> Dict[Tuple[int, int], KgInfo]:
    """Synthetic platform inventory matching gen.spans.SpanGenConfig: every
    (epc, ip) the generator can emit gets pod/node/service ids (the
    controller-lite pushes this on startup in benchmarks/tests)."""
    entries: Dict[Tuple[int, int], KgInfo] = {}
    for ipl in range(cfg.n_ips):
        ip = 0x0A000000 | ipl
        epc = 1 + (ip % cfg.n_epcs)
        entries[(epc, ip)] = KgInfo(
            pod_id=1 + ipl,
            pod_node_id=1 + ipl % 64,
            pod_ns_id=1 + ipl % 16,
            pod_group_id=1 + ipl % 512,
            pod_cluster_id=1,
            l3_device_type=14,  # pod device type
            l3_device_id=1 + ipl,
            subnet_id=1 + ipl % 32,
            host_id=1 + ipl % 64,
            az_id=1 + ipl % 4,
            service_id=1 + ipl % 256,
            gprocess_id=1 + ipl % 1024,
        )
    return entries
