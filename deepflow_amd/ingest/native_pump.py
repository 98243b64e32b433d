"""Native receiver fast path: C++ per-connection pump threads.

The general `Receiver` (receiver.py) handles every message type in
Python — fine for control-plane volumes.  The flow-log data plane is
another matter: at millions of spans/s the Python deframe/decompress
loop is the bottleneck (~0.6 GB/s/stream measured).  `PumpServer`
accepts agent connections and hands each socket to a native pump
(ops/csrc/recv_pump.cpp): recv -> trident deframe -> zstd -> a pinned
SPSC byte ring, zero interpreter work per frame.  The consumer walks
the ring with numpy views over pinned memory and feeds slices straight
to the GPU pipeline (hipMemcpyAsync sees pinned pages — no staging
copy).

Reference counterpart: server/libs/receiver/receiver.go flow
per-connection goroutines + hash-to-queue dispatch.
"""
from __future__ import annotations

import ctypes as ct
import os
import socket
import threading
import time
from typing import Callable, List, Optional

import numpy as np
import torch

from ..ops import native
from ..wire import framing

WRAP_MARK = (1 << 64) - 1


class NativePump:
    """One native pump bound to one connected socket.

    The payload ring is a pinned torch uint8 tensor; entries are
    [u64 len][payload][pad-to-8].  `poll()` yields zero-copy numpy views
    into the ring; call `advance()` once a view's bytes are consumed
    (after the H2D copy completes for GPU consumers).
    """

    def __init__(self, sock: socket.socket, ring_bytes: int = 256 << 20,
                 accept_type: int = framing.MSG_PROTOCOLLOG,
                 pin: Optional[bool] = None):
        if pin is None:
            pin = torch.cuda.is_available()
        self.lib = native.cpu()
        self.ring_t = torch.empty(ring_bytes, dtype=torch.uint8,
                                  pin_memory=pin)
        self.ring = self.ring_t.numpy()
        self.cap = ring_bytes
        fd = os.dup(sock.fileno())
        sock.close()
        self.h = self.lib.df_pump_start(
            fd, ct.c_void_p(self.ring_t.data_ptr()), self.cap,
            accept_type)
        self.tail = 0

    def poll(self) -> Optional[np.ndarray]:
        """Next payload view, or None if the ring is empty.  The view
        aliases pinned ring memory — valid until `advance()`; frame
        metadata for it is in self.meta (msg_type, agent_id, org_id,
        team_id)."""
        while True:
            head = self.lib.df_pump_head(self.h)
            if head == self.tail:
                return None
            pos = self.tail % self.cap
            hdr = self.ring[pos:pos + 16].view(np.uint64)
            ln = int(hdr[0])
            if ln == WRAP_MARK:
                self.tail += self.cap - pos
                continue
            m = int(hdr[1])
            self.meta = (m & 0xFF, (m >> 8) & 0xFFFF, (m >> 24) & 0xFFFF,
                         (m >> 40) & 0xFFFFFFFF)
            self._entry = 16 + ((ln + 7) & ~7)
            return self.ring[pos + 16: pos + 16 + ln]

    def advance(self) -> None:
        self.tail += self._entry
        self.lib.df_pump_set_tail(self.h, self.tail)

    def pending(self) -> int:
        return self.lib.df_pump_head(self.h) - self.tail

    def done(self) -> bool:
        return bool(self.lib.df_pump_done(self.h)) and self.pending() == 0

    def stats(self) -> dict:
        v = [ct.c_uint64() for _ in range(4)]
        self.lib.df_pump_stats(self.h, *(ct.byref(x) for x in v))
        return {"frames": v[0].value, "wire_bytes": v[1].value,
                "payload_bytes": v[2].value, "bad_frames": v[3].value}

    def close(self) -> None:
        if self.h:
            self.lib.df_pump_free(self.h)
            self.h = None


class PumpServer:
    """Accept loop that spawns a NativePump per connection and runs a
    consumer thread draining every ring.

    handler(payload_view, meta) is called with each frame's decoded
    payload (numpy view over pinned memory) and its frame metadata
    (msg_type, agent_id, org_id, team_id); it must finish consuming the bytes
    (or enqueue an async copy and return an event-like with .query())
    before the view is recycled.  If the handler returns an object with
    a query() method, the pump defers the ring advance until it reports
    True — this is how the GPU path keeps hipMemcpyAsync in flight
    without racing the producer.
    """

    def __init__(self, handler: Callable[[np.ndarray, tuple], object],
                 host: str = "127.0.0.1", port: int = 0,
                 ring_bytes: int = 256 << 20,
                 accept_type: int = framing.MSG_PROTOCOLLOG,
                 pin: Optional[bool] = None,
                 idle_handler: Optional[Callable[[], None]] = None):
        self.handler = handler
        self.idle_handler = idle_handler
        self.ring_bytes = ring_bytes
        self.accept_type = accept_type
        self.pin = pin
        self.pumps: List[NativePump] = []
        self._pending: List[tuple] = []  # (pump, event) awaiting H2D
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._sock.bind((host, port))
        self.port = self._sock.getsockname()[1]
        self._sock.listen(64)
        self._sock.settimeout(0.2)
        self._threads = [
            threading.Thread(target=self._accept_loop, daemon=True),
            threading.Thread(target=self._consume_loop, daemon=True),
        ]

    def start(self) -> "PumpServer":
        for t in self._threads:
            t.start()
        return self

    def _accept_loop(self) -> None:
        while not self._stop.is_set():
            try:
                conn, _ = self._sock.accept()
            except socket.timeout:
                continue
            except OSError:
                return
            p = NativePump(conn, self.ring_bytes, self.accept_type,
                           self.pin)
            with self._lock:
                self.pumps.append(p)

    def _consume_loop(self) -> None:
        while not self._stop.is_set():
            busy = False
            with self._lock:
                pumps = list(self.pumps)
            # retire completed async copies first so their ring space
            # frees before the producers stall
            still = []
            for p, ev in self._pending:
                if ev.query():
                    p.advance()
                else:
                    still.append((p, ev))
            self._pending = still
            for p in pumps:
                if any(q is p for q, _ in self._pending):
                    continue  # strictly in-order per pump
                view = p.poll()
                if view is None:
                    continue
                busy = True
                ev = self.handler(view, p.meta)
                if ev is not None and hasattr(ev, "query"):
                    self._pending.append((p, ev))
                else:
                    p.advance()
            if not busy and not self._pending:
                if self.idle_handler is not None:
                    self.idle_handler()
                time.sleep(0.0005)

    def stats(self) -> dict:
        with self._lock:
            pumps = list(self.pumps)
        agg = {"frames": 0, "wire_bytes": 0, "payload_bytes": 0,
               "bad_frames": 0, "connections": len(pumps)}
        for p in pumps:
            for k, v in p.stats().items():
                agg[k] += v
        return agg

    def stop(self) -> None:
        self._stop.set()
        for t in self._threads:
            t.join(timeout=2)
        self._sock.close()
        with self._lock:
            for p in self.pumps:
                p.close()
            self.pumps.clear()


class GpuL7Feeder:
    """PumpServer handler feeding the GPU L7 pipeline with coalesced
    ingests: frames' pinned views H2D-copy into one device aggregation
    buffer, record offsets rebase onto it, and ONE ingest runs per
    ~`agg_spans` spans (per-frame ingests of ~90k spans are kernel-
    launch bound).  Returns a CUDA event per frame so the PumpServer
    recycles ring space only after the copies complete.  Use the bound
    methods as PumpServer(handler=feeder, idle_handler=feeder.idle):
    `idle` flushes a partial aggregation once the wire goes quiet."""

    def __init__(self, pipe, max_records: int = 1 << 21,
                 agg_spans: int = 1_000_000, agg_bytes: int = 512 << 20,
                 ingest_lock=None):
        import ctypes as ct
        self.pipe = pipe
        # serializes the coalesced ingest against concurrent queries
        # when the owner (the server) shares state with an engine
        self.ingest_lock = ingest_lock
        self.max_records = max_records
        self.agg_spans = agg_spans
        self.agg_bytes = agg_bytes
        self.lib = native.cpu()
        self._ct = ct
        # 6-deep rotation: each pair's async H2D completes in us — deep
        # enough to never overwrite an in-flight copy, shallow enough to
        # keep pinned memory bounded (max_records=2^21 -> 96 MB)
        self._scratch = [(torch.empty(max_records, dtype=torch.int32,
                                      pin_memory=True),
                          torch.empty(max_records, dtype=torch.int32,
                                      pin_memory=True))
                         for _ in range(6)]
        self._si = 0
        self._buf = torch.empty(agg_bytes, dtype=torch.uint8,
                                device="cuda")
        self._offs = torch.empty(4 * max_records, dtype=torch.int32,
                                 device="cuda")
        self._lens = torch.empty(4 * max_records, dtype=torch.int32,
                                 device="cuda")
        self._used = 0
        self._n = 0
        self._last_append = 0.0
        self._lock = threading.Lock()

    def _flush_locked(self) -> None:
        if self._n == 0:
            return
        if self.ingest_lock is not None:
            with self.ingest_lock:
                self.pipe.ingest_device(self._buf[: self._used],
                                        self._offs[: self._n],
                                        self._lens[: self._n],
                                        self._buf[: self._used])
        else:
            self.pipe.ingest_device(self._buf[: self._used],
                                    self._offs[: self._n],
                                    self._lens[: self._n],
                                    self._buf[: self._used])
        self._used = 0
        self._n = 0

    def __call__(self, view: np.ndarray, meta: tuple):
        ct = self._ct
        with self._lock:
            offs_p, lens_p = self._scratch[self._si % len(self._scratch)]
            self._si += 1
            n = int(self.lib.df_scan_offsets(
                ct.c_void_p(view.ctypes.data), len(view),
                ct.c_void_p(offs_p.data_ptr()),
                ct.c_void_p(lens_p.data_ptr()), self.max_records))
            if self._used + len(view) > self.agg_bytes or                     self._n + n > 4 * self.max_records:
                self._flush_locked()
            base = self._used
            self._buf[base: base + len(view)].copy_(
                torch.from_numpy(view), non_blocking=True)
            offs_np = offs_p.numpy()
            offs_np[:n] += base
            self._offs[self._n: self._n + n].copy_(offs_p[:n],
                                                   non_blocking=True)
            self._lens[self._n: self._n + n].copy_(lens_p[:n],
                                                   non_blocking=True)
            self._used += (len(view) + 7) & ~7
            self._n += n
            self._last_append = time.monotonic()
            ev = torch.cuda.Event()
            ev.record()
            if self._n >= self.agg_spans:
                self._flush_locked()
            return ev

    def idle(self) -> None:
        """Wire quiet: flush a partial aggregation after a short dwell
        (PumpServer idle_handler)."""
        if self._n and time.monotonic() - self._last_append > 0.02:
            with self._lock:
                self._flush_locked()
