"""UDP debug command bus (reference: server/libs/debug udp.go — the
`deepflow-ctl ingester` transport): newline-JSON request/response over a
localhost datagram socket, serving live counters, store stats and queue
depths without touching the HTTP data plane.
"""
from __future__ import annotations

import json
import socket
import threading
from typing import Callable, Dict, Optional


class DebugBus:
    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self.handlers: Dict[str, Callable[[dict], object]] = {}
        self.sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        self.sock.bind((host, port))
        self.port = self.sock.getsockname()[1]
        self.sock.settimeout(0.5)
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def register(self, cmd: str, fn: Callable[[dict], object]) -> None:
        self.handlers[cmd] = fn

    def _serve(self) -> None:
        while not self._stop.is_set():
            try:
                data, addr = self.sock.recvfrom(1 << 16)
            except socket.timeout:
                continue
            except OSError:
                return
            try:
                req = json.loads(data.decode())
                fn = self.handlers.get(req.get("cmd", ""))
                if fn is None:
                    resp = {"error": f"unknown cmd {req.get('cmd')!r}",
                            "cmds": sorted(self.handlers)}
                else:
                    resp = {"result": fn(req)}
            except Exception as e:  # noqa: BLE001 — bus must survive junk
                resp = {"error": str(e)}
            try:
                self.sock.sendto(json.dumps(resp, default=str).encode(),
                                 addr)
            except OSError:
                pass

    def start(self) -> None:
        self._thread = threading.Thread(target=self._serve, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)
        self.sock.close()


def debug_call(port: int, cmd: str, timeout: float = 2.0, **kw) -> dict:
    """Client side (deepflow-ctl ingester analog)."""
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    s.settimeout(timeout)
    try:
        s.sendto(json.dumps({"cmd": cmd, **kw}).encode(),
                 ("127.0.0.1", port))
        data, _ = s.recvfrom(1 << 16)
        return json.loads(data.decode())
    finally:
        s.close()
