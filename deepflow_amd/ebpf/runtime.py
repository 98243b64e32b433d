"""EbpfCollector: SkEvents -> socket resolution -> agent FlowMap.

Reference counterpart: agent/src/ebpf_dispatcher.rs:460-520 (SK_BPF_DATA
-> MetaPacket -> FlowMap). The kernel program ships (tgid, fd) instead of
kernel-struct socket pointers (no CO-RE — see progs.py); this collector
resolves (tgid, fd) -> 4-tuple from /proc and feeds the payloads into
agent_core's dfa_syscall_batch, where the SAME L7 parsers as the packet
path produce AppProtoLogsData with signal_source=EBPF and
syscall_trace_id request/response join keys.

Event sources: the kernel perf buffer (loader.SocketTracer.poll), the
userspace VM (tests), or recorded event bytes (replay files).
"""
from __future__ import annotations

import os
import struct
from typing import Callable, Dict, List, Optional, Tuple

from .progs import EV_HDR, SK_EVENT_FMT

Tuple4 = Tuple[int, int, int, int, int]  # ip_src, ip_dst, sport, dport, proto


class ProcSocketResolver:
    """(tgid, fd) -> TCP/UDP 4-tuple via /proc/<pid>/fd + /proc/net/tcp.

    The inode map refreshes lazily on miss; entries cache forever (a
    given (tgid, fd, inode) binding never changes its 4-tuple)."""

    def __init__(self, proc_root: str = "/proc"):
        self.proc = proc_root
        self._by_inode: Dict[int, Tuple4] = {}
        self._cache: Dict[Tuple[int, int], Tuple4] = {}

    def _refresh_inodes(self) -> None:
        for name, proto in (("tcp", 6), ("udp", 17)):
            path = f"{self.proc}/net/{name}"
            try:
                with open(path) as f:
                    next(f)
                    for line in f:
                        parts = line.split()
                        if len(parts) < 10:
                            continue
                        lip, lport = parts[1].split(":")
                        rip, rport = parts[2].split(":")
                        inode = int(parts[9])
                        # /proc/net addresses are LE hex
                        self._by_inode[inode] = (
                            int.from_bytes(bytes.fromhex(lip), "little"),
                            int.from_bytes(bytes.fromhex(rip), "little"),
                            int(lport, 16), int(rport, 16), proto)
            except OSError:
                pass

    def resolve(self, tgid: int, fd: int) -> Optional[Tuple4]:
        key = (tgid, fd)
        hit = self._cache.get(key)
        if hit is not None:
            return hit
        try:
            link = os.readlink(f"{self.proc}/{tgid}/fd/{fd}")
        except OSError:
            return None
        if not link.startswith("socket:["):
            return None
        inode = int(link[8:-1])
        if inode not in self._by_inode:
            self._refresh_inodes()
        tup = self._by_inode.get(inode)
        if tup is not None:
            self._cache[key] = tup
        return tup


class StaticResolver:
    """Test/replay resolver: explicit (tgid, fd) -> tuple map."""

    def __init__(self, table: Dict[Tuple[int, int], Tuple4]):
        self.table = dict(table)

    def resolve(self, tgid: int, fd: int) -> Optional[Tuple4]:
        return self.table.get((tgid, fd))


# dfa_syscall_batch record: [ts u64][tgid u32][dir u8][proto u8][hint u8]
# [pad u8][ip_src u32][ip_dst u32][psrc u16][pdst u16][trace u64][len u32]
_REC_FMT = "<QIBBBxIIHHQI"
assert struct.calcsize(_REC_FMT) == 40


TLS_FD = 0xFFFFFFFF


class EbpfCollector:
    def __init__(self, agent, resolver=None,
                 drop_stale_s: "Optional[float]" = None):
        """drop_stale_s: in LIVE pumping (sub-100ms processing lag),
        drop events older than this — the ring backlog from before the
        workload would otherwise resolve against CURRENT /proc state
        and attribute a recycled fd number's old traffic to the new
        socket. Leave None for replay/batch feeding."""
        self.agent = agent
        self.resolver = resolver or ProcSocketResolver()
        self.drop_stale_s = drop_stale_s
        self.stale_dropped = 0
        self.events_in = 0
        self.unresolved = 0
        self._batch: List[bytes] = []
        # TLS pairing: SSL_read/SSL_write uprobe events carry no fd (no
        # portable way to read it from the SSL object without struct
        # offsets) — pair them with the thread's most recent socket
        # syscall, the reference's fallback when offsets are unknown
        self._tid_last_fd: Dict[Tuple[int, int], int] = {}

    def on_event(self, ev: bytes) -> None:
        """One SkEvent (perf record payload) from any source."""
        if len(ev) < EV_HDR:
            return
        (ts, tgid, pid, fd, ln, cap, direction, proto_hint, _sc, trace,
         _skey) = struct.unpack(SK_EVENT_FMT, ev[:EV_HDR])
        if self.drop_stale_s is not None:
            import time as _time
            now = _time.clock_gettime(_time.CLOCK_MONOTONIC) * 1e9
            if ts < now - self.drop_stale_s * 1e9:
                self.stale_dropped += 1
                return
        payload = ev[EV_HDR:EV_HDR + cap]
        if fd == TLS_FD:
            fd = self._tid_last_fd.get((tgid, pid), -1)
        else:
            # pair TLS-uprobe events only with sockets whose syscalls
            # carry TLS ciphertext (record header 0x14-0x17 0x03 0x0x) —
            # "last fd of the thread" alone grabs unrelated chatty fds
            if len(payload) >= 3 and 0x14 <= payload[0] <= 0x17 and \
                    payload[1] == 0x03 and payload[2] <= 0x04:
                self._tid_last_fd[(tgid, pid)] = fd
                if len(self._tid_last_fd) > (1 << 16):
                    self._tid_last_fd.clear()
        tup = self.resolver.resolve(tgid, fd) if fd >= 0 else None
        self.events_in += 1
        if tup is None:
            self.unresolved += 1
            return
        ip_s, ip_d, p_s, p_d, proto = tup
        self._batch.append(struct.pack(
            _REC_FMT, ts, tgid, direction, proto, proto_hint,
            ip_s, ip_d, p_s, p_d, trace, len(payload)) + payload)

    def flush(self) -> int:
        if not self._batch:
            return 0
        blob = b"".join(self._batch)
        self._batch.clear()
        return self.agent.syscall_batch(blob)

    # ------------------------------------------------------------ replay
    def replay(self, blob: bytes) -> int:
        """Recorded perf-event stream: [u32 size][SkEvent bytes]..."""
        pos = 0
        n = 0
        while pos + 4 <= len(blob):
            (size,) = struct.unpack_from("<I", blob, pos)
            pos += 4
            self.on_event(blob[pos:pos + size])
            pos += size
            n += 1
        self.flush()
        return n


def record_events(events: List[bytes]) -> bytes:
    """Serialize SkEvents into the replay file format."""
    out = bytearray()
    for e in events:
        out += struct.pack("<I", len(e)) + e
    return bytes(out)
