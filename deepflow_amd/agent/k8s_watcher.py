"""Kubernetes ApiWatcher: pod/node/service inventory via the watch API.

Reference counterpart: agent/src/platform/kubernetes/api_watcher.rs +
resource_watcher.rs — the agent LISTs then WATCHes the apiserver and
reports the inventory through genesis so the controller can build
PlatformData/KnowledgeGraph entries.

Speaks the plain k8s REST protocol (no client library): an initial
LIST (`GET /api/v1/<kind>`), then a WATCH from the returned
resourceVersion (`?watch=true&resourceVersion=N`) reading JSON-lines
events. Inventory diffs flow to a sink callback as
{(epc, ip) -> KgInfo-ish dict} updates — the server wires this into
ControllerLite.update_platform (tests run it against a fake apiserver).
"""
from __future__ import annotations

import ipaddress
import json
import socket
import threading
import urllib.request
from typing import Callable, Dict, Optional, Tuple


def _ip_u32(ip: str) -> int:
    try:
        return int(ipaddress.IPv4Address(ip))
    except (ipaddress.AddressValueError, ValueError):
        return 0


class K8sApiWatcher:
    """Watches pods (extensible to nodes/services) on one apiserver."""

    def __init__(self, api_url: str, sink: Callable[[Dict], None],
                 epc_id: int = 1, token: Optional[str] = None,
                 kinds: Tuple[str, ...] = ("pods",)):
        self.api_url = api_url.rstrip("/")
        self.sink = sink
        self.epc_id = epc_id
        self.token = token
        self.kinds = kinds
        self.pods: Dict[str, Dict] = {}   # uid -> summary
        self.events = 0
        self.resyncs = 0
        self._stop = threading.Event()
        self._threads = []

    # ------------------------------------------------------------ http
    def _get(self, path: str, stream: bool = False):
        req = urllib.request.Request(self.api_url + path)
        if self.token:
            req.add_header("Authorization", f"Bearer {self.token}")
        return urllib.request.urlopen(req, timeout=30 if stream else 10)

    # ------------------------------------------------------------ model
    def _pod_entry(self, obj: Dict) -> Optional[Tuple[str, Dict]]:
        meta = obj.get("metadata", {})
        status = obj.get("status", {})
        ip = status.get("podIP", "")
        if not ip:
            return None
        uid = meta.get("uid", meta.get("name", ""))
        return uid, {
            "name": meta.get("name", ""),
            "namespace": meta.get("namespace", "default"),
            "node": obj.get("spec", {}).get("nodeName", ""),
            "ip": _ip_u32(ip),
            "labels": meta.get("labels", {}),
        }

    def _emit(self, entry: Dict, deleted: bool = False) -> None:
        self.sink({
            "kind": "pod",
            "deleted": deleted,
            "epc": self.epc_id,
            **entry,
        })

    def _handle_event(self, ev: Dict) -> None:
        etype = ev.get("type")
        obj = ev.get("object", {})
        pe = self._pod_entry(obj)
        if pe is None:
            return
        uid, entry = pe
        self.events += 1
        if etype in ("ADDED", "MODIFIED"):
            self.pods[uid] = entry
            self._emit(entry)
        elif etype == "DELETED":
            self.pods.pop(uid, None)
            self._emit(entry, deleted=True)

    # ------------------------------------------------------------ loops
    def list_once(self) -> str:
        """Initial LIST; emits everything; returns resourceVersion."""
        with self._get("/api/v1/pods") as resp:
            body = json.load(resp)
        for obj in body.get("items", []):
            pe = self._pod_entry(obj)
            if pe:
                uid, entry = pe
                self.pods[uid] = entry
                self._emit(entry)
        return body.get("metadata", {}).get("resourceVersion", "0")

    def watch_once(self, rv: str) -> None:
        """One WATCH connection: JSON-lines events until EOF."""
        path = f"/api/v1/pods?watch=true&resourceVersion={rv}"
        with self._get(path, stream=True) as resp:
            for line in resp:
                if self._stop.is_set():
                    return
                line = line.strip()
                if not line:
                    continue
                try:
                    self._handle_event(json.loads(line))
                except json.JSONDecodeError:
                    continue

    def _run(self) -> None:
        while not self._stop.is_set():
            try:
                rv = self.list_once()
                self.watch_once(rv)
            except (OSError, socket.timeout, json.JSONDecodeError):
                self.resyncs += 1
            self._stop.wait(1.0)

    def start(self) -> None:
        t = threading.Thread(target=self._run, daemon=True)
        t.start()
        self._threads.append(t)

    def stop(self) -> None:
        self._stop.set()
        for t in self._threads:
            t.join(timeout=3)


def platform_entries_from_pods(events, epc_default: int = 1):
    """Fold watcher events into {(epc, ip) -> KgInfo} platform updates +
    name maps (the genesis->controller shape)."""
    from ..store.kg import KgInfo
    entries: Dict[Tuple[int, int], "KgInfo"] = {}
    names: Dict[str, Dict[int, str]] = {"pod_map": {}}
    for i, ev in enumerate(events):
        if ev.get("kind") != "pod" or ev.get("deleted"):
            continue
        pod_id = (hash(ev["name"]) & 0x7FFFFF) or 1
        entries[(ev.get("epc", epc_default), ev["ip"])] = KgInfo(
            pod_id=pod_id,
            pod_ns_id=(hash(ev["namespace"]) & 0xFFFF) or 1,
            pod_node_id=(hash(ev["node"]) & 0xFFFF) or 1,
        )
        names["pod_map"][pod_id] = ev["name"]
    return entries, names
