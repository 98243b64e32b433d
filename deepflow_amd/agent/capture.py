"""AF_PACKET capture dispatcher: live frames -> the C++ flow engine.

The reference's dispatcher/recv_engine family (af_packet/DPDK) feeds
FlowMap from a kernel ring; here a (optionally fanout-sharded) AF_PACKET
SOCK_RAW socket per worker drains frames straight into `dfa_packet`.
PACKET_FANOUT_HASH gives the same flow-affinity sharding the reference
relies on: one flow always lands on the same worker's FlowMap.
"""
from __future__ import annotations

import socket
import struct
import threading
import time
from typing import List, Optional

ETH_P_ALL = 0x0003
SOL_PACKET = 263
PACKET_FANOUT = 18
PACKET_FANOUT_HASH = 0


class CaptureWorker:
    def __init__(self, agent, iface: str = "lo",
                 fanout_group: Optional[int] = None,
                 snaplen: int = 65535):
        self.agent = agent
        self.iface = iface
        self.snaplen = snaplen
        self.packets = 0
        self.bytes = 0
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.sock = socket.socket(socket.AF_PACKET, socket.SOCK_RAW,
                                  socket.htons(ETH_P_ALL))
        self.sock.bind((iface, 0))
        self.sock.settimeout(0.2)
        if fanout_group is not None:
            # flow-hash fanout: all packets of a flow go to one worker
            opt = struct.pack("<HH", fanout_group, PACKET_FANOUT_HASH)
            self.sock.setsockopt(SOL_PACKET, PACKET_FANOUT, opt)

    def _loop(self) -> None:
        recv = self.sock.recvfrom
        packet = self.agent.packet
        while not self._stop.is_set():
            try:
                frame, addr = recv(self.snaplen)
            except socket.timeout:
                continue
            except OSError:
                return
            # addr = (iface, proto, pkttype, hatype, hwaddr);
            # pkttype 4 = outgoing — keep both directions
            self.packets += 1
            self.bytes += len(frame)
            packet(frame, time.time_ns())

    def start(self) -> None:
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
        self.sock.close()


class CaptureDispatcher:
    """N fanout workers, each with its own Agent flow engine (the
    reference's per-dispatcher FlowMap layout). For n_workers == 1 the
    single agent needs no fanout group."""

    def __init__(self, agents: List, iface: str = "lo"):
        group = (id(self) & 0xFFFF) if len(agents) > 1 else None
        self.workers = [CaptureWorker(a, iface, fanout_group=group)
                        for a in agents]

    def start(self) -> None:
        for w in self.workers:
            w.start()

    def stop(self) -> None:
        for w in self.workers:
            w.stop()

    def stats(self) -> dict:
        return {"packets": sum(w.packets for w in self.workers),
                "bytes": sum(w.bytes for w in self.workers),
                "workers": len(self.workers)}
