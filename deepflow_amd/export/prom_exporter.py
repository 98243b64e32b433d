"""Prometheus exporter: rollup metrics in text exposition format.

Reference counterpart: server/ingester/exporters (prometheus exporter) —
re-exports enriched flow metrics for an external Prometheus to scrape.
Serves the flow_metrics family tables as
`deepflow_<table>_<field>{tag="..."} value` on /metrics.
"""
from __future__ import annotations

from typing import Dict, Iterable, List


def _esc(v) -> str:
    return str(v).replace("\\", "\\\\").replace('"', '\\"').replace(
        "\n", "\\n")


def render_rows(prefix: str, rows: Iterable[Dict], fields: List[str],
                out: List[str]) -> None:
    for row in rows:
        labels = ",".join(
            f'{k}="{_esc(v)}"' for k, v in sorted(row.items())
            if k not in fields and k != "time")
        ts_ms = row.get("time", 0) * 1000
        for f in fields:
            if f in row:
                out.append(f"{prefix}_{f}{{{labels}}} {row[f]} {ts_ms}")


class PromExporter:
    """Renders the newest rollup window on scrape (stateless)."""

    def __init__(self, l7_pipeline=None, l4_pipeline=None,
                 max_series: int = 20000):
        self.l7 = l7_pipeline
        self.l4 = l4_pipeline
        self.max_series = max_series

    def render(self) -> str:
        from ..store.metrics import APP_FIELDS, NET_FIELDS
        out: List[str] = []
        if self.l7 is not None:
            t = self.l7.rollups.get("application.1m")
            out.append("# TYPE deepflow_application_request counter")
            render_rows("deepflow_application", t.rows()[-self.max_series:],
                        APP_FIELDS, out)
        if self.l4 is not None:
            t = self.l4.rollups.get("network.1m")
            out.append("# TYPE deepflow_network_byte_tx counter")
            render_rows("deepflow_network", t.rows()[-self.max_series:],
                        NET_FIELDS, out)
        return "\n".join(out) + "\n"

    def register(self, app) -> None:
        from fastapi.responses import PlainTextResponse

        @app.get("/metrics", response_class=PlainTextResponse)
        def metrics():
            return self.render()
