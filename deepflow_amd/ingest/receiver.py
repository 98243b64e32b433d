"""Trident-protocol receiver: TCP + UDP servers on :20033.

Accepts the reference agent's framed protobuf streams (19-byte header,
SURVEY.md appendix A), tracks per-agent status, decompresses zstd payloads,
and dispatches payloads to registered per-message-type handlers.
Reference counterpart: server/libs/receiver/receiver.go.
"""
from __future__ import annotations

import ctypes as ct
import queue as _queue
import socket
import struct
import threading
from typing import Callable, Dict, Optional, Tuple

import numpy as np

from ..utils.stats import Counter
from ..wire import framing
from ..ops import native

Handler = Callable[[framing.FrameHeader, np.ndarray], None]
# handler receives (header, payload uint8 ndarray)

MAX_FRAME = 64 << 20


def _decompress(encoder: int, payload: bytes) -> Optional[np.ndarray]:
    if encoder == framing.ENCODER_RAW:
        return np.frombuffer(payload, dtype=np.uint8)
    if encoder == framing.ENCODER_ZSTD:
        lib = native.cpu()
        src = np.frombuffer(payload, dtype=np.uint8)
        cap = max(len(payload) * 20, 1 << 20)
        while cap <= MAX_FRAME * 64:
            dst = np.zeros(cap, dtype=np.uint8)
            n = lib.df_zstd_decompress(src.ctypes.data, len(src),
                                       dst.ctypes.data, cap)
            if n >= 0:
                return dst[:n]
            # retry with a larger buffer: highly repetitive payloads can
            # exceed the 20x first-guess ratio
            cap *= 4
        return None
    return None


class AgentStatus:
    """Per-(agent, msg_type) sequence/status accounting
    (receiver.go:199-318 equivalent)."""

    def __init__(self):
        self.frames = 0
        self.bytes = 0
        self.last_seen = 0.0
        self.decode_errors = 0


class Receiver:
    def __init__(self, tcp_port: int = 20033, udp_port: int = 20033,
                 host: str = "127.0.0.1", queue_depth: int = 256):
        self.host = host
        self.tcp_port = tcp_port
        self.udp_port = udp_port
        self.handlers: Dict[int, Handler] = {}
        self.status: Dict[Tuple[int, int], AgentStatus] = {}
        self.counter = Counter("ingester.receiver")
        self._stop = threading.Event()
        self._threads = []
        self._tcp_sock: Optional[socket.socket] = None
        self._udp_sock: Optional[socket.socket] = None
        # decode queue decoupling socket reads from pipeline work
        # (reference: receiver hashes frames to N decoder queues with
        # overwrite-on-full drop accounting, receiver.go:519-566)
        self._queue: "_queue.Queue" = _queue.Queue(maxsize=queue_depth)
        self._dispatcher: Optional[threading.Thread] = None

    def register(self, msg_type: int, handler: Handler) -> None:
        self.handlers[msg_type] = handler

    # ------------------------------------------------------------- frames
    def handle_frame(self, frame: bytes) -> bool:
        """Process one complete frame (also the in-process entry point used
        by all-in-one mode and tests)."""
        try:
            hdr, payload, _ = framing.decode_frame(frame)
        except ValueError:
            self.counter.add("invalid_frames")
            return False
        st = self.status.setdefault((hdr.agent_id, hdr.msg_type), AgentStatus())
        st.frames += 1
        st.bytes += len(frame)
        handler = self.handlers.get(hdr.msg_type)
        if handler is None:
            self.counter.add("unhandled_type")
            return False
        data = _decompress(hdr.encoder, payload)
        if data is None:
            st.decode_errors += 1
            self.counter.add("decompress_errors")
            return False
        self.counter.add("frames_in")
        self.counter.add("bytes_in", len(frame))
        handler(hdr, data)
        return True

    def enqueue_frame(self, frame: bytes) -> bool:
        """Queue a frame for the dispatcher thread; drops (with counter)
        when the pipeline is backed up — at-most-once like the reference's
        overwrite queues."""
        if self._dispatcher is None:
            return self.handle_frame(frame)
        try:
            self._queue.put_nowait(frame)
            return True
        except _queue.Full:
            self.counter.add("queue_drops")
            return False

    def _dispatch_loop(self) -> None:
        while not self._stop.is_set():
            try:
                frame = self._queue.get(timeout=0.5)
            except _queue.Empty:
                continue
            try:
                self.handle_frame(frame)
            except Exception:  # noqa: BLE001 — a bad frame must not kill
                self.counter.add("handler_errors")

    # ------------------------------------------------------------- servers
    def start(self) -> None:
        self._dispatcher = threading.Thread(target=self._dispatch_loop,
                                            daemon=True)
        self._dispatcher.start()
        self._threads.append(self._dispatcher)
        self._tcp_sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._tcp_sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._tcp_sock.bind((self.host, self.tcp_port))
        self.tcp_port = self._tcp_sock.getsockname()[1]
        self._tcp_sock.listen(64)
        self._tcp_sock.settimeout(0.5)
        t = threading.Thread(target=self._tcp_loop, daemon=True)
        t.start()
        self._threads.append(t)

        self._udp_sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        self._udp_sock.bind((self.host, self.udp_port if self.udp_port else 0))
        self.udp_port = self._udp_sock.getsockname()[1]
        self._udp_sock.settimeout(0.5)
        t2 = threading.Thread(target=self._udp_loop, daemon=True)
        t2.start()
        self._threads.append(t2)

    def stop(self) -> None:
        self._stop.set()
        for t in self._threads:
            t.join(timeout=2)
        if self._tcp_sock:
            self._tcp_sock.close()
        if self._udp_sock:
            self._udp_sock.close()

    def _tcp_loop(self) -> None:
        while not self._stop.is_set():
            try:
                conn, _ = self._tcp_sock.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            t = threading.Thread(target=self._conn_loop, args=(conn,),
                                 daemon=True)
            t.start()
            # daemon connection threads exit with their sockets; keeping
            # references forever would leak under connection churn
            self._threads = [th for th in self._threads if th.is_alive()]
            self._threads.append(t)

    def _conn_loop(self, conn: socket.socket) -> None:
        conn.settimeout(1.0)
        # bytearray accumulation (amortized append) + deferred compaction:
        # bytes-concat reassembly copied the whole partial frame per 1 MB
        # chunk (~16x write amplification on 32 MB frames)
        buf = bytearray()
        pos = 0
        while not self._stop.is_set():
            try:
                chunk = conn.recv(1 << 20)
            except socket.timeout:
                continue
            except OSError:
                break
            if not chunk:
                break
            buf += chunk
            while len(buf) - pos >= 4:
                (size,) = struct.unpack_from(">I", buf, pos)
                if size > MAX_FRAME or size < framing.HEADER_LEN:
                    self.counter.add("invalid_frames")
                    buf = bytearray()
                    pos = 0
                    break
                if len(buf) - pos < size:
                    break
                self.enqueue_frame(bytes(buf[pos:pos + size]))
                pos += size
            if pos and (pos == len(buf) or pos > (8 << 20)):
                del buf[:pos]
                pos = 0
        conn.close()

    def _udp_loop(self) -> None:
        while not self._stop.is_set():
            try:
                data, _ = self._udp_sock.recvfrom(1 << 16)
            except socket.timeout:
                continue
            except OSError:
                return
            if len(data) >= framing.HEADER_LEN:
                self.enqueue_frame(data)
