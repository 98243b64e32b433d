"""Process/socket scanner — the genesis data source.

Reference counterpart: agent platform_synchronizer (linux_process.rs +
socket scanner) feeding controller/genesis: the agent reports local
processes and listening/established sockets; the controller assigns global
process ids (GPIDs) that the data plane stamps into flow logs
(gpid_0/gpid_1), which is what makes cross-host process-to-process tracing
joinable. (The eBPF tracer is the round-2 source for per-socket data; this
scanner is the /proc-based path that also runs without eBPF.)
"""
from __future__ import annotations

import os
from typing import Dict, List


def scan_processes(limit: int = 2000) -> List[Dict]:
    out = []
    for entry in sorted(os.listdir("/proc")):
        if not entry.isdigit():
            continue
        pid = int(entry)
        try:
            with open(f"/proc/{pid}/comm") as f:
                comm = f.read().strip()
            with open(f"/proc/{pid}/cmdline", "rb") as f:
                cmdline = f.read().replace(b"\0", b" ").decode(
                    "utf-8", "replace").strip()
            stat = os.stat(f"/proc/{pid}")
            out.append({"pid": pid, "name": comm, "cmdline": cmdline[:256],
                        "uid": stat.st_uid})
        except (OSError, PermissionError):
            continue
        if len(out) >= limit:
            break
    return out


def _parse_proc_net_tcp(path: str) -> List[Dict]:
    out = []
    try:
        with open(path) as f:
            lines = f.read().splitlines()[1:]
    except OSError:
        return out
    for line in lines:
        parts = line.split()
        if len(parts) < 10:
            continue
        def addr(s):
            host, port = s.rsplit(":", 1)
            if len(host) == 8:  # ipv4 little-endian hex
                ip = int(host, 16)
                ip = ((ip & 0xFF) << 24) | ((ip & 0xFF00) << 8) | \
                     ((ip >> 8) & 0xFF00) | (ip >> 24)
            else:
                ip = 0  # ipv6 collapsed for the inventory
            return ip, int(port, 16)
        lip, lport = addr(parts[1])
        rip, rport = addr(parts[2])
        out.append({"local_ip": lip, "local_port": lport,
                    "remote_ip": rip, "remote_port": rport,
                    "state": int(parts[3], 16),
                    "inode": int(parts[9])})
    return out


def scan_sockets() -> List[Dict]:
    socks = _parse_proc_net_tcp("/proc/net/tcp")
    # map socket inodes to pids (bounded scan of /proc/*/fd)
    inode_to_pid: Dict[int, int] = {}
    scanned = 0
    for entry in os.listdir("/proc"):
        if not entry.isdigit():
            continue
        if scanned > 400:
            break
        pid = int(entry)
        scanned += 1
        try:
            for fd in os.listdir(f"/proc/{pid}/fd"):
                try:
                    link = os.readlink(f"/proc/{pid}/fd/{fd}")
                except OSError:
                    continue
                if link.startswith("socket:["):
                    inode_to_pid[int(link[8:-1])] = pid
        except OSError:
            continue
    for s in socks:
        s["pid"] = inode_to_pid.get(s["inode"], 0)
    return socks


class GenesisReporter:
    """Periodically pushes the local inventory to the controller and keeps
    the returned pid -> GPID map for the data plane."""

    def __init__(self, agent_id: int, post_fn):
        """post_fn(payload_dict) -> response_dict (HTTP or in-process)."""
        self.agent_id = agent_id
        self.post = post_fn
        self.gpids: Dict[int, int] = {}

    def report(self) -> Dict[int, int]:
        payload = {
            "agent_id": self.agent_id,
            "processes": scan_processes(),
            "sockets": scan_sockets(),
        }
        resp = self.post(payload)
        self.gpids = {int(k): v for k, v in resp.get("gpids", {}).items()}
        return self.gpids
