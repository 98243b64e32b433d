"""Cloud/Kubernetes inventory pollers feeding the knowledge graph.

The reference's cloud recorder polls provider APIs (aws/k8s/...) and
writes resource tables that tagrecorder turns into id->name maps
(server/controller/cloud, server/controller/recorder). Here a poller
periodically pulls a snapshot from a source callable (a cluster API
client, a file exported by an external collector, or a test fixture),
normalizes it to (epc, ip) -> KgInfo entries + name maps, diffs against
the last snapshot, and pushes only changes into ControllerLite
(version-gated, so agents and the GPU KG join table update together).
"""
from __future__ import annotations

import ipaddress
import threading
import time
from typing import Callable, Dict, Optional, Tuple

from ..store.kg import KgInfo


def k8s_snapshot_to_platform(snap: Dict) -> Tuple[Dict, Dict]:
    """Normalize a k8s-style snapshot {pods: [...], nodes: [...],
    services: [...]} into (platform entries, name maps)."""
    entries: Dict[Tuple[int, int], KgInfo] = {}
    names: Dict[str, Dict[int, str]] = {"pod": {}, "pod_node": {},
                                        "pod_ns": {}, "service": {},
                                        "pod_cluster": {}}
    ns_ids: Dict[str, int] = {}
    for i, pod in enumerate(snap.get("pods", []), start=1):
        ip = int(ipaddress.IPv4Address(pod["ip"]))
        epc = pod.get("epc", 0)
        ns = pod.get("namespace", "default")
        ns_id = ns_ids.setdefault(ns, len(ns_ids) + 1)
        pod_id = pod.get("id", i)
        entries[(epc, ip)] = KgInfo(
            pod_id=pod_id, pod_ns_id=ns_id,
            pod_node_id=pod.get("node_id", 0),
            pod_cluster_id=snap.get("cluster_id", 1),
            pod_group_id=pod.get("group_id", 0))
        names["pod"][pod_id] = pod["name"]
        names["pod_ns"][ns_id] = ns
    for node in snap.get("nodes", []):
        names["pod_node"][node["id"]] = node["name"]
        if "ip" in node:
            ip = int(ipaddress.IPv4Address(node["ip"]))
            entries[(node.get("epc", 0), ip)] = KgInfo(
                pod_node_id=node["id"],
                pod_cluster_id=snap.get("cluster_id", 1))
    for svc in snap.get("services", []):
        names["service"][svc["id"]] = svc["name"]
        if "cluster_ip" in svc:
            ip = int(ipaddress.IPv4Address(svc["cluster_ip"]))
            entries[(svc.get("epc", 0), ip)] = KgInfo(
                service_id=svc["id"],
                pod_cluster_id=snap.get("cluster_id", 1))
    if "cluster_id" in snap:
        names["pod_cluster"][snap["cluster_id"]] = \
            snap.get("cluster_name", f"cluster-{snap['cluster_id']}")
    return entries, names


class CloudPoller:
    """Periodic snapshot poller with change detection."""

    def __init__(self, controller, source: Callable[[], Dict],
                 normalize: Callable[[Dict], Tuple[Dict, Dict]] =
                 k8s_snapshot_to_platform,
                 interval_s: float = 30.0):
        self.controller = controller
        self.source = source
        self.normalize = normalize
        self.interval_s = interval_s
        self.last_entries: Optional[Dict] = None
        self.polls = 0
        self.pushes = 0
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def poll_once(self) -> bool:
        """Pull + normalize + push-if-changed. Returns True on a push."""
        snap = self.source()
        self.polls += 1
        entries, names = self.normalize(snap)
        if entries == self.last_entries:
            return False
        self.controller.update_platform(entries, names=names)
        self.last_entries = entries
        self.pushes += 1
        return True

    def start(self) -> None:
        def loop():
            while not self._stop.is_set():
                try:
                    self.poll_once()
                except Exception:  # noqa: BLE001 — a flaky source must
                    pass           # not kill the poller
                self._stop.wait(self.interval_s)
        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)


def filereader_source(path: str) -> Callable[[], Dict]:
    """The reference's `filereader` cloud provider
    (controller/cloud/filereader): the inventory lives in a YAML/JSON
    file an operator (or external exporter) maintains; each poll reads
    the file and the normal snapshot-diff machinery applies changes.
    Supports the k8s-style snapshot shape plus a flat `hosts:` list of
    {name, ip, epc, host_id, az}."""
    import json
    import os

    def load() -> Dict:
        with open(path) as f:
            text = f.read()
        if path.endswith((".yaml", ".yml")):
            import yaml
            snap = yaml.safe_load(text) or {}
        else:
            snap = json.loads(text or "{}")
        return snap
    return load


def filereader_snapshot_to_platform(snap: Dict) -> Tuple[Dict, Dict]:
    """Normalize a filereader inventory: k8s-shaped sections reuse the
    k8s normalizer; a flat `hosts:` section maps to host/az entries."""
    entries, names = k8s_snapshot_to_platform(snap)
    names.setdefault("host", {})
    names.setdefault("az", {})
    az_ids: Dict[str, int] = {}
    for i, h in enumerate(snap.get("hosts", []), start=1):
        ip = int(ipaddress.IPv4Address(h["ip"]))
        az = h.get("az", "")
        az_id = az_ids.setdefault(az, len(az_ids) + 1) if az else 0
        host_id = h.get("host_id", i)
        entries[(h.get("epc", 0), ip)] = KgInfo(
            host_id=host_id, az_id=az_id,
            l3_device_id=h.get("device_id", host_id), l3_device_type=2)
        names["host"][host_id] = h["name"]
        if az:
            names["az"][az_id] = az
    return entries, names


class FileReaderProvider(CloudPoller):
    """CloudPoller over a local inventory file (reference filereader)."""

    def __init__(self, controller, path: str, interval_s: float = 30.0):
        super().__init__(controller, filereader_source(path),
                         normalize=filereader_snapshot_to_platform,
                         interval_s=interval_s)
