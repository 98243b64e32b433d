"""Python control wrapper for the C++ agent core (ops/csrc/agent_core.cpp)
plus the uniform sender (framing + TCP to the ingester).

Reference counterparts: agent/src/trident.rs component lifecycle,
sender/uniform_sender.rs. Packet sources: synthetic frames in tests;
AF_PACKET capture on a deployment host feeds dfa_packet the same way.
"""
from __future__ import annotations

import ctypes as ct
import socket
from typing import Dict, Optional

import numpy as np

import struct

from ..ops import native
from ..wire import framing

DRAIN_L4, DRAIN_L7, DRAIN_DOC, DRAIN_PCAP, DRAIN_NPB = 0, 1, 2, 3, 4

_MSG_FOR = {DRAIN_L4: framing.MSG_TAGGEDFLOW,
            DRAIN_L7: framing.MSG_PROTOCOLLOG,
            DRAIN_DOC: framing.MSG_METRICS,
            DRAIN_PCAP: framing.MSG_RAW_PCAP}


def _lib():
    lib = native.cpu()
    if not hasattr(lib, "_agent_decl"):
        p, u32, u64 = ct.c_void_p, ct.c_uint32, ct.c_uint64
        lib.dfa_new.restype = p
        lib.dfa_new.argtypes = [u32]
        lib.dfa_free.argtypes = [p]
        lib.dfa_add_cidr.argtypes = [p, u32, u32, ct.c_int32]
        lib.dfa_add_acl.argtypes = [p, u32, u32, u32, u32, u32, u32, u32,
                                    u32, u32]
        lib.dfa_add_custom_port.argtypes = [p, u32]
        lib.dfa_packet.restype = ct.c_int
        lib.dfa_packet.argtypes = [p, p, u32, u64]
        lib.dfa_packet_batch.restype = ct.c_int64
        lib.dfa_packet_batch.argtypes = [p, p, u64]
        lib.dfa_syscall_batch.restype = ct.c_int64
        lib.dfa_syscall_batch.argtypes = [p, p, u64]
        lib.dfa_tick.argtypes = [p, u64]
        lib.dfa_drain.restype = u64
        lib.dfa_drain.argtypes = [p, ct.c_int, p, u64]
        lib.dfa_stats.argtypes = [p, p]
        lib._agent_decl = True
    return lib


class Agent:
    def __init__(self, vtap_id: int = 1, agent_id: Optional[int] = None,
                 server: Optional[tuple] = None, team_id: int = 0,
                 org_id: int = 1):
        self._lib = _lib()
        self._h = self._lib.dfa_new(vtap_id)
        self.vtap_id = vtap_id
        self.agent_id = agent_id if agent_id is not None else vtap_id
        self.team_id = team_id
        self.org_id = org_id
        self.server = server
        self._sock: Optional[socket.socket] = None
        self._server_idx = 0

    def close(self) -> None:
        if self._h:
            self._lib.dfa_free(self._h)
            self._h = None
        if self._sock:
            self._sock.close()
            self._sock = None

    def add_cidr(self, net: int, masklen: int, epc: int) -> None:
        self._lib.dfa_add_cidr(self._h, net, masklen, epc)

    def add_acl(self, gid: int, src_net: int = 0, src_masklen: int = 0,
                dst_net: int = 0, dst_masklen: int = 0, proto: int = 0,
                port_min: int = 0, port_max: int = 65535,
                action: int = 0) -> None:
        """FlowAcl rule (policy/labeler analog): matched once per new
        flow, cached on the flow, gids emitted in TaggedFlow acl_gids."""
        self._lib.dfa_add_acl(self._h, gid, src_net, src_masklen, dst_net,
                              dst_masklen, proto, port_min, port_max,
                              action)

    def add_custom_protocol_port(self, port: int) -> None:
        """Port-rule custom protocol (L7 proto 127); raw sessions are
        captured and re-parsed by plugins (agent/plugins.py)."""
        self._lib.dfa_add_custom_port(self._h, port)

    def register_plugin(self, parse_fn) -> None:
        from .plugins import PluginHost
        if not hasattr(self, "plugin_host"):
            self.plugin_host = PluginHost()
        self.plugin_host.register(parse_fn)

    def packet(self, frame: bytes, ts_ns: int) -> int:
        buf = np.frombuffer(frame, dtype=np.uint8)
        return self._lib.dfa_packet(self._h, buf.ctypes.data, len(frame),
                                    ts_ns)

    def packet_batch(self, blob: bytes) -> int:
        """Batch entry: [u32 len][u64 ts][frame]... packed; one native
        call for the whole batch (capture rings, pps benches)."""
        buf = np.frombuffer(blob, dtype=np.uint8)
        return int(self._lib.dfa_packet_batch(self._h, buf.ctypes.data,
                                              len(blob)))

    def syscall_batch(self, blob: bytes) -> int:
        """eBPF socket-trace events (ebpf/runtime.py wire format) into
        the FlowMap + L7 parsers; one native call per drained batch."""
        buf = np.frombuffer(blob, dtype=np.uint8)
        return int(self._lib.dfa_syscall_batch(self._h, buf.ctypes.data,
                                               len(blob)))

    def start_ebpf(self, poll_interval_s: float = 0.05):
        """Attach the socket tracer (raw bpf(2), ebpf/loader.py) and pump
        its perf events into the FlowMap on a background thread. Returns
        the tracer, or None where BPF is unavailable (the guard/melt-down
        path: packet capture keeps running — reference disables eBPF on
        old kernels the same way, utils/guard.rs:778-788)."""
        from ..ebpf import loader, runtime
        if not loader.available():
            return None
        tracer = loader.SocketTracer()
        tracer.attach()
        coll = runtime.EbpfCollector(self, drop_stale_s=1.0)
        import threading
        stop = threading.Event()

        def pump():
            while not stop.wait(poll_interval_s):
                batch: list = []
                tracer.poll(batch.append)
                # per-cpu rings drain ring-by-ring: sort the batch by
                # event timestamp so TLS-uprobe events see the pairing
                # syscall first (EV_TS is the leading u64)
                batch.sort(key=lambda e: int.from_bytes(e[:8], 'little'))
                for ev in batch:
                    coll.on_event(ev)
                coll.flush()
        th = threading.Thread(target=pump, daemon=True)
        th.start()
        tracer._stop = stop  # noqa: SLF001 — owned here
        self._ebpf = (tracer, coll, th)
        return tracer

    def stop_ebpf(self) -> None:
        eb = getattr(self, "_ebpf", None)
        if eb:
            tracer, _coll, _th = eb
            tracer._stop.set()
            tracer.close()
            self._ebpf = None

    def tick(self, now_ns: int) -> None:
        self._lib.dfa_tick(self._h, now_ns)

    def drain(self, which: int) -> bytes:
        n = self._lib.dfa_drain(self._h, which, None, 0)
        if n == 0:
            return b""
        out = np.zeros(int(n), dtype=np.uint8)
        got = self._lib.dfa_drain(self._h, which, out.ctypes.data, n)
        payload = out[:got].tobytes()
        if which == DRAIN_L7 and getattr(self, "plugin_host", None):
            payload = self.plugin_host.process_l7_payload(payload)
        return payload

    def stats(self) -> Dict[str, int]:
        arr = np.zeros(8, dtype=np.uint64)
        self._lib.dfa_stats(self._h, arr.ctypes.data)
        keys = ["flows_active", "flows_emitted", "l7_emitted", "docs_emitted",
                "packets", "bytes", "parse_errors", "_r"]
        return dict(zip(keys, (int(x) for x in arr)))

    # ---------------------------------------------------------- sender
    def frame(self, which: int, payload: bytes,
              compress: bool = False) -> bytes:
        encoder = framing.ENCODER_RAW
        if compress and len(payload) > 128:
            payload_z = self._zstd(payload)
            if payload_z is not None and len(payload_z) < len(payload):
                payload = payload_z
                encoder = framing.ENCODER_ZSTD
        hdr = framing.FrameHeader(msg_type=_MSG_FOR[which],
                                  team_id=self.team_id, org_id=self.org_id,
                                  agent_id=self.agent_id, encoder=encoder)
        return framing.encode_frame(hdr, payload)

    def _zstd(self, payload: bytes):
        src = np.frombuffer(payload, dtype=np.uint8)
        dst = np.zeros(len(payload) + 1024, dtype=np.uint8)
        n = self._lib.df_zstd_compress(src.ctypes.data, len(src),
                                       dst.ctypes.data, len(dst), 3)
        return dst[:n].tobytes() if n > 0 else None

    # ---------------------------------------------------------- guard
    # exception bits (reference agent/src/exception.rs bitmask idea)
    EXC_MEM_LIMIT = 1 << 0
    EXC_FLOW_TABLE_FULL = 1 << 1
    EXC_SERVER_UNREACHABLE = 1 << 2

    def guard_check(self, max_memory_mb: int = 768,
                    max_flows: int = 1 << 20) -> int:
        """Melt-down circuit breaker (reference utils/guard.rs:770-790):
        returns the exception bitmask; callers stop feeding packets when
        non-zero (the reference disables dispatchers)."""
        exc = 0
        try:
            with open("/proc/self/statm") as f:
                rss_pages = int(f.read().split()[1])
            rss_mb = rss_pages * 4096 // (1 << 20)
            if rss_mb > max_memory_mb:
                exc |= self.EXC_MEM_LIMIT
        except OSError:
            pass
        if self.stats()["flows_active"] > max_flows:
            exc |= self.EXC_FLOW_TABLE_FULL
        self.exceptions = exc
        return exc

    # escape timer: if the controller stays unreachable past escape_s,
    # the agent self-disables its capture (reference
    # rpc/synchronizer.rs:1464 escape behavior)
    ESCAPE_S_DEFAULT = 3600

    def escaped(self, now_s: float, escape_s: Optional[float] = None) -> bool:
        last = getattr(self, "last_sync_ok", None)
        if last is None:
            return False  # never synced: standalone mode
        return (now_s - last) > (escape_s or self.ESCAPE_S_DEFAULT)

    # ---------------------------------------------------------- sync
    def sync_with_controller(self, post_fn) -> dict:
        """One trident-Synchronizer cycle: report versions + exceptions,
        apply pushed config and platform data (CIDR->EPC for the labeler).
        post_fn(dict) -> dict (HTTP client or in-process)."""
        resp = post_fn({
            "agent_id": self.agent_id,
            "hostname": socket.gethostname(),
            "config_version": getattr(self, "config_version", 0),
            "platform_version": getattr(self, "platform_version", 0),
            "exceptions": getattr(self, "exceptions", 0),
        })
        if "config" in resp:
            self.config = resp["config"]
            self.config_version = resp["config_version"]
        if "platform" in resp:
            for entry in resp["platform"]:
                # epc labeling for every known endpoint (/32)
                self.add_cidr(entry["ip"], 32, entry["epc"])
            self.platform_version = resp["platform_version"]
        import time as _time
        self.last_sync_ok = _time.time()
        return resp

    def dfstats_payload(self, now_s: int = 0) -> bytes:
        """Agent self-metrics as a MSG_DFSTATS payload (reference: the
        agent's stats collector shipping deepflow_system rows)."""
        from ..wire import pb, metric
        st = self.stats()
        rec = {
            "timestamp": now_s,
            "name": "deepflow_agent",
            "tag_names": ["host", "agent_id"],
            "tag_values": [socket.gethostname(), str(self.agent_id)],
            "metrics_float_names": [k for k in st if k != "_r"],
            "metrics_float_values": [float(st[k]) for k in st if k != "_r"],
            "org_id": self.org_id,
            "team_id": self.team_id,
        }
        return framing.pack_records([pb.encode(rec, metric.STATS)])

    def set_npb_target(self, host: str, port: int = 4789) -> None:
        """Configure the NPB (north-bound packet broker) tunnel target;
        ACL-matched frames with the NPB action (bit1) are mirrored as
        VXLAN/UDP datagrams (reference: agent handler/npb.rs)."""
        import socket as _socket
        self._npb_target = (host, port)
        self._npb_sock = _socket.socket(_socket.AF_INET,
                                        _socket.SOCK_DGRAM)

    def flush_npb(self) -> int:
        """Drain the NPB mirror buffer and ship one VXLAN datagram per
        mirrored frame. Returns datagrams sent."""
        if getattr(self, "_npb_sock", None) is None:
            return 0
        buf = self.drain(DRAIN_NPB)
        sent = 0
        pos = 0
        while pos + 2 <= len(buf):
            (ln,) = struct.unpack_from("<H", buf, pos)
            pos += 2
            if pos + ln > len(buf):
                break
            try:
                self._npb_sock.sendto(buf[pos:pos + ln], self._npb_target)
                sent += 1
            except OSError:
                pass
            pos += ln
        return sent

    def flush_to_server(self, now_ns: int, compress: bool = False,
                        with_stats: bool = True) -> int:
        """tick + drain all types + send framed payloads to the server
        (uniform-sender analog; compress=True uses whole-payload zstd like
        the reference's SenderEncoder::Zstd). Also ships a MSG_DFSTATS
        self-metrics frame. Returns frames sent."""
        self.tick(now_ns)
        self.flush_npb()
        frames = []
        for which in (DRAIN_L4, DRAIN_L7, DRAIN_DOC, DRAIN_PCAP):
            payload = self.drain(which)
            if payload:
                frames.append(self.frame(which, payload, compress=compress))
        if with_stats:
            hdr = framing.FrameHeader(msg_type=framing.MSG_DFSTATS,
                                      team_id=self.team_id,
                                      org_id=self.org_id,
                                      agent_id=self.agent_id)
            frames.append(framing.encode_frame(
                hdr, self.dfstats_payload(now_ns // 10**9)))
        for frame in frames:
            if self.server is not None:
                self._send_with_failover(frame)
        return len(frames)

    def _send_with_failover(self, frame: bytes, retries: int = 2) -> None:
        """uniform-sender failover: on send failure, reconnect — rotating
        through the server list when more than one ingester is known."""
        servers = self.server if isinstance(self.server, list) \
            else [self.server]
        last: Optional[Exception] = None
        for attempt in range(retries + 1):
            try:
                if self._sock is None:
                    target = servers[(self._server_idx + attempt)
                                     % len(servers)]
                    self._sock = socket.create_connection(target, timeout=5)
                    self._server_idx = (self._server_idx + attempt) % \
                        len(servers)
                self._sock.sendall(frame)
                return
            except OSError as e:
                last = e
                if self._sock:
                    self._sock.close()
                self._sock = None
        self.exceptions = getattr(self, "exceptions", 0) | \
            self.EXC_SERVER_UNREACHABLE
        raise ConnectionError(f"all ingesters unreachable: {last}")
