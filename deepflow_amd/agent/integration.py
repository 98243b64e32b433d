"""Agent integration collector: local HTTP push endpoints relayed to the
server over the trident protocol.

Reference counterpart: agent/src/integration_collector.rs (HTTP :38086):
  POST /api/v1/otel/trace       OTLP traces (pb; zlib'd before relay)
  POST /api/v1/prometheus       prometheus remote-write (WriteRequest pb)
  POST /api/v1/profile/ingest   pyroscope-style folded profile push
  POST /api/v1/telegraf         influx line protocol (relayed as app logs)
"""
from __future__ import annotations

import socket
import zlib
from typing import Optional, Tuple

from fastapi import FastAPI, Request

from ..wire import pb, metric, framing


class IntegrationCollector:
    def __init__(self, server: Tuple[str, int], agent_id: int = 1,
                 org_id: int = 1, team_id: int = 0):
        self.server = server
        self.agent_id = agent_id
        self.org_id = org_id
        self.team_id = team_id
        self._sock: Optional[socket.socket] = None
        self.relayed = {"otel": 0, "prometheus": 0, "profile": 0,
                        "telegraf": 0}
        self.app = self._build()

    def _send(self, msg_type: int, payload: bytes) -> None:
        hdr = framing.FrameHeader(msg_type=msg_type, agent_id=self.agent_id,
                                  org_id=self.org_id, team_id=self.team_id)
        frame = framing.encode_frame(hdr, payload)
        if self._sock is None:
            self._sock = socket.create_connection(self.server, timeout=5)
        self._sock.sendall(frame)

    def _build(self) -> FastAPI:
        app = FastAPI(title="deepflow-amd agent integration")

        @app.post("/api/v1/otel/trace")
        async def otel_trace(request: Request):
            body = await request.body()
            if request.headers.get("content-encoding") == "gzip":
                import gzip
                body = gzip.decompress(body)
            # relay zlib-compressed, as the reference agent does
            # (integration_collector.rs compresses OTLP before relay)
            self._send(framing.MSG_OPENTELEMETRY, zlib.compress(body))
            self.relayed["otel"] += 1
            return {"status": "ok"}

        @app.post("/api/v1/prometheus")
        async def prometheus(request: Request):
            body = await request.body()
            self._send(framing.MSG_PROMETHEUS, body)
            self.relayed["prometheus"] += 1
            return {"status": "ok"}

        @app.post("/api/v1/profile/ingest")
        async def profile_ingest(request: Request):
            body = await request.body()
            params = dict(request.query_params)
            prof = {
                "name": params.get("name", "external"),
                "units": params.get("units", "samples"),
                "format": params.get("format", "folded"),
                "spy_name": params.get("spyName", "external"),
                "data": body,
                "from_time": int(params.get("from", "0") or 0),
                "until": int(params.get("until", "0") or 0),
                "event_type": 0,
            }
            payload = framing.pack_records([pb.encode(prof, metric.PROFILE)])
            self._send(framing.MSG_PROFILE, payload)
            self.relayed["profile"] += 1
            return {"status": "ok"}

        @app.post("/api/v1/telegraf")
        async def telegraf(request: Request):
            body = await request.body()
            self._send(framing.MSG_APPLICATION_LOG, body)
            self.relayed["telegraf"] += 1
            return {"status": "ok"}

        @app.get("/api/v1/status")
        def status():
            return self.relayed

        return app
