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


TPACKET_V3 = 2
PACKET_VERSION = 10
PACKET_RX_RING = 5
TP_STATUS_USER = 1


class RingCapture:
    """TPACKET_V3 mmap block-ring capture (reference:
    dispatcher/recv_engine/af_packet/tpacket.rs). The kernel fills
    block-sized chunks of a shared ring; each ready block drains with ONE
    native call (dfa_ring_block walks the tpacket3 chain straight into
    the C++ flow engine) — no per-packet syscalls, no per-packet Python.
    """

    def __init__(self, agent, iface: str = "lo",
                 block_size: int = 1 << 20, block_nr: int = 64,
                 fanout_group: Optional[int] = None,
                 retire_tov_ms: int = 60):
        import ctypes as ct
        import mmap
        self.agent = agent
        self.packets = 0
        self.blocks = 0
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.block_size = block_size
        self.block_nr = block_nr
        self.sock = socket.socket(socket.AF_PACKET, socket.SOCK_RAW,
                                  socket.htons(ETH_P_ALL))
        self.sock.setsockopt(SOL_PACKET, PACKET_VERSION, TPACKET_V3)
        # tpacket_req3: block_size, block_nr, frame_size, frame_nr,
        # retire_blk_tov, sizeof_priv, feature_req_word
        frame_size = 2048
        frame_nr = block_size // frame_size * block_nr
        req = struct.pack("<7I", block_size, block_nr, frame_size,
                          frame_nr, retire_tov_ms, 0, 0)
        self.sock.setsockopt(SOL_PACKET, PACKET_RX_RING, req)
        self.sock.bind((iface, 0))
        if fanout_group is not None:
            opt = struct.pack("<HH", fanout_group, PACKET_FANOUT_HASH)
            self.sock.setsockopt(SOL_PACKET, PACKET_FANOUT, opt)
        self.mm = mmap.mmap(self.sock.fileno(), block_size * block_nr,
                            mmap.MAP_SHARED,
                            mmap.PROT_READ | mmap.PROT_WRITE)
        self._buf = (ct.c_char * (block_size * block_nr)).from_buffer(
            self.mm)
        self._base = ct.addressof(self._buf)
        self._lib = agent._lib
        if not hasattr(self._lib, "_ring_decl"):
            self._lib.dfa_ring_block.restype = ct.c_int64
            self._lib.dfa_ring_block.argtypes = [ct.c_void_p, ct.c_void_p]
            self._lib._ring_decl = True

    def _drain_block(self, i: int) -> int:
        import ctypes as ct
        base = self._base + i * self.block_size
        status = ct.c_uint32.from_address(base + 8)
        if not (status.value & TP_STATUS_USER):
            return 0
        n = int(self._lib.dfa_ring_block(self.agent._h, base))
        # release the block back to the kernel
        status.value = 0
        self.packets += n
        self.blocks += 1
        return n

    def poll_once(self, timeout_ms: int = 100) -> int:
        import select
        p = select.poll()
        p.register(self.sock, select.POLLIN)
        p.poll(timeout_ms)
        drained = 0
        for i in range(self.block_nr):
            drained += self._drain_block(i)
        return drained

    def _loop(self) -> None:
        cur = 0
        import select
        poller = select.poll()
        poller.register(self.sock, select.POLLIN)
        while not self._stop.is_set():
            n = self._drain_block(cur)
            if n:
                cur = (cur + 1) % self.block_nr
                continue
            poller.poll(50)
        # final sweep
        for i in range(self.block_nr):
            self._drain_block(i)

    def start(self) -> None:
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
        del self._buf
        self.mm.close()
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
