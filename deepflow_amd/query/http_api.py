"""Querier HTTP API (reference: server/querier/router, port 20416).

POST /v1/query/            form/json {db, sql} -> {columns, values}
GET  /v1/health
GET  /v1/stats             self-telemetry snapshot
Tempo + PromQL routes are registered by their apps (tempo.py / promql.py).
"""
from __future__ import annotations

from typing import Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse


def build_app(engine, registry=None, tempo=None, promql=None,
              profile=None, tracing=None, engine_for=None) -> FastAPI:
    app = FastAPI(title="deepflow-amd querier")

    @app.get("/v1/health")
    def health():
        return {"status": "ok"}

    @app.post("/v1/query/")
    async def query(request: Request):
        ctype = request.headers.get("content-type", "")
        # org isolation: X-Org-Id header routes to that org's engine
        eng = engine
        org_hdr = request.headers.get("x-org-id")
        if org_hdr and engine_for is not None:
            try:
                eng = engine_for(int(org_hdr))
            except ValueError:
                pass
        sql: Optional[str] = None
        db = "flow_log"
        if "json" in ctype:
            body = await request.json()
            sql = body.get("sql")
            db = body.get("db", db)
        else:
            # parse urlencoded form without python-multipart
            from urllib.parse import parse_qs
            raw = (await request.body()).decode("utf-8", "replace")
            form = {k: v[0] for k, v in parse_qs(raw).items()}
            sql = form.get("sql")
            db = form.get("db", db)
        if not sql:
            return JSONResponse({"OPT_STATUS": "INVALID_PARAMETERS",
                                 "DESCRIPTION": "missing sql"}, status_code=400)
        try:
            result = eng.query(sql)
        except Exception as e:  # noqa: BLE001
            return JSONResponse({"OPT_STATUS": "FAILED",
                                 "DESCRIPTION": str(e)}, status_code=400)
        return {"OPT_STATUS": "SUCCESS",
                "result": {"columns": result["columns"],
                           "values": result["values"]}}

    @app.get("/v1/stats")
    def stats():
        if registry is None:
            return []
        return registry.snapshot_all()

    if tempo is not None:
        tempo.register(app)
    if tracing is not None:
        tracing.register(app)
    if promql is not None:
        promql.register(app)
    if profile is not None:
        profile.register(app)
    return app
