"""All-in-one server: receiver + GPU ingest pipelines + query engine + HTTP.

The MI355X-native analog of the reference's single server binary running
ingester + querier (+ controller-lite) in one process
(server/cmd/server/main.go:108-117). One process per GPU; rank/shard wiring
comes from deepflow_amd.parallel.
"""
from __future__ import annotations

import threading
from typing import Optional

import numpy as np

from fastapi import Request

from .gen import SpanGenConfig
from .ingest import L7IngestPipeline
from .ingest.receiver import Receiver
from .query import QueryEngine
from .query.http_api import build_app
from .store.kg import KnowledgeGraphTable, default_platform
from .utils.stats import default_registry
from .wire import framing


class DeepflowServer:
    def __init__(self, device: str = "cpu", tcp_port: int = 0,
                 segment_rows: int = 1 << 20,
                 dict_capacity: int = 1 << 20,
                 time_base_s: int = 1_700_000_000,
                 platform_cfg: Optional[SpanGenConfig] = None,
                 native_pump: bool = False,
                 pump_port: int = 0):
        self.device = device
        self.kg = KnowledgeGraphTable(device=device)
        if platform_cfg is not None:
            self.kg.update(default_platform(platform_cfg))
        self.l7 = L7IngestPipeline(device=device, segment_rows=segment_rows,
                                   kg=self.kg, dict_capacity=dict_capacity,
                                   time_base_s=time_base_s)
        from .ingest.l4_pipeline import L4IngestPipeline
        self.l4 = L4IngestPipeline(device=device, segment_rows=segment_rows,
                                   kg=self.kg, time_base_s=time_base_s)
        self.receiver = Receiver(tcp_port=tcp_port, udp_port=0)
        self.receiver.register(framing.MSG_PROTOCOLLOG, self._on_l7)
        self.receiver.register(framing.MSG_TAGGEDFLOW, self._on_l4)
        self.engine = QueryEngine(self.l7, device=device, l4_pipeline=self.l4)
        # native data-plane receiver (ops/csrc/recv_pump.cpp): C++
        # deframe/zstd straight into a pinned ring, dispatched into the
        # same per-type handlers. Started on its own port by start()
        # when native_pump=True (agents point their data plane at it;
        # the Python receiver keeps serving control-plane traffic).
        self._use_pump = native_pump
        self._pump_port = pump_port
        self.pump = None
        self.pump_port = None
        # multi-org isolation: each non-default org gets its own KG,
        # dictionary, segment sets and engine (reference: per-org
        # ClickHouse databases, org_id from the frame header / ORG_ID
        # HTTP header). Aux pipelines (logs/events/profiles) stay global.
        self.default_org = 1
        self._org_ctx: dict = {}
        from .query.tempo import TempoApp
        from .query.tracing import DistributedTracer
        self.tempo = TempoApp(self.engine)
        self.tracer = DistributedTracer(self.engine)
        from .query.promql import PromQLEngine
        from .ingest.prom_pipeline import PromPipeline
        from .ingest.doc_pipeline import DocPipeline
        self.docs = DocPipeline()
        self.receiver.register(framing.MSG_METRICS,
                               lambda hdr, payload:
                               self.docs.ingest_payload(payload.tobytes()))
        self.prom = PromPipeline(device=device)
        self.receiver.register(framing.MSG_PROMETHEUS,
                               lambda hdr, payload:
                               self.prom.ingest_write_request(
                                   payload.tobytes()))
        self.promql = PromQLEngine(self.l7.metrics.rows, self.l4.metrics.rows,
                                   raw_sources=[self.prom.series_for],
                                   sql_engine=self.engine)
        from .ingest.profile_pipeline import ProfilePipeline, ProfileApp
        self.profiles = ProfilePipeline()
        self.receiver.register(framing.MSG_PROFILE,
                               lambda hdr, payload:
                               self.profiles.ingest_payload(payload.tobytes()))
        self.receiver.register(framing.MSG_OPENTELEMETRY, self._on_otel)
        self.receiver.register(framing.MSG_OPENTELEMETRY_COMPRESSED,
                               self._on_otel)
        self.receiver.register(
            framing.MSG_SKYWALKING,
            lambda hdr, payload: self._on_third_party(hdr, payload,
                                                      "skywalking"))
        self.receiver.register(
            framing.MSG_DATADOG,
            lambda hdr, payload: self._on_third_party(hdr, payload,
                                                      "datadog"))
        from .ingest.pcap_pipeline import PcapPipeline
        self.pcap = PcapPipeline()
        self.receiver.register(framing.MSG_RAW_PCAP,
                               lambda hdr, payload:
                               self.pcap.ingest_payload(payload.tobytes()))
        self.receiver.register(framing.MSG_PACKETSEQUENCE,
                               lambda hdr, payload:
                               self.pcap.ingest_payload(payload.tobytes()))
        from .ingest.event_pipeline import EventPipeline
        from .ingest.applog_pipeline import AppLogPipeline
        self.events = EventPipeline()
        self.applogs = AppLogPipeline()
        self.receiver.register(framing.MSG_PROC_EVENT,
                               lambda hdr, payload:
                               self.events.ingest_proc_events(
                                   payload.tobytes()))
        for mt in (framing.MSG_APPLICATION_LOG, framing.MSG_SYSLOG,
                   framing.MSG_AGENT_LOG):
            self.receiver.register(
                mt, lambda hdr, payload:
                self.applogs.ingest_lines(payload.tobytes(),
                                          agent_id=hdr.agent_id))
        from .control import ControllerLite
        self.controller = ControllerLite(
            kg=self.kg, event_sink=self.events.add_resource_event)
        # controller-global persistent prometheus label ids (reference
        # GetPrometheusLabelIDs): rebuild the pipeline's interners over
        # the allocator now that the controller exists
        from .ingest.prom_pipeline import GlobalInterner
        self.prom.metric_names = GlobalInterner(
            "metric", self.controller.alloc_prom_ids)
        self.prom.label_names = GlobalInterner(
            "label_name", self.controller.alloc_prom_ids)
        self.prom.label_values = GlobalInterner(
            "label_value", self.controller.alloc_prom_ids)
        self.system_rows = []  # deepflow_system self-metrics store
        self.receiver.register(framing.MSG_DFSTATS, self._on_dfstats)
        self.engine.system_rows = self.system_rows
        # tagrecorder hydration: the engine resolves KG ids to display
        # names through the controller's name maps (live reference)
        self.engine.name_maps = self.controller.name_maps
        self.engine.agent_app_rows = lambda: self.docs.app_rows
        self.engine.agent_net_rows = lambda: self.docs.net_rows
        self.engine.event_rows = lambda: self.events.resource_events
        self.engine.perf_event_rows = lambda: self.events.perf_events
        self.engine.alert_event_rows = lambda: self.events.alert_events
        self.engine.app_log_rows = lambda: self.applogs.rows
        self.engine.trace_tree_rows = lambda: self.tracer.tree_rows
        self.app = build_app(self.engine, registry=default_registry(),
                             tempo=self.tempo, tracing=self.tracer,
                             promql=self.promql,
                             profile=ProfileApp(self.profiles),
                             engine_for=self.engine_for_org)
        self.controller.register(self.app)
        from .export import OtlpExporter
        self.exporter = OtlpExporter(self.engine)
        from .export.prom_exporter import PromExporter
        self.prom_exporter = PromExporter(self.l7, self.l4)
        self.prom_exporter.register(self.app)
        from .query.mcp import McpServer
        self.mcp = McpServer(self.engine, self.profiles)
        self.mcp.register(self.app)

        @self.app.get("/v1/debug/threads")
        def debug_threads():
            # self-tracing aid (reference: server self-pprof,
            # ingester/droplet/profiler)
            import sys
            import traceback
            out = {}
            for tid, frame in sys._current_frames().items():
                out[str(tid)] = traceback.format_stack(frame)[-4:]
            return out

        @self.app.get("/v1/debug/store")
        def debug_store():
            segs = self.l7.segments
            return {
                "l7_rows": segs.n_rows,
                "l7_segments": len(segs.segments),
                "l7_alloc_bytes": segs.total_alloc_bytes(),
                "l7_evicted_rows": segs.evicted_rows,
                "layout_version": getattr(segs.segments[0],
                                          "layout_version", 0)
                if segs.segments else 0,
                "dict_entries": self.l7.dict.n_entries(),
                "l4_rows": self.l4.segments.n_rows,
            }

        @self.app.post("/v1/debug/self-profile")
        def self_profile(seconds: float = 1.0):
            # continuous self-profiling hook: capture our own GPU kernel
            # activity into the profile store (reference: server pushes its
            # own pprof into DeepFlow, cmd/server/config.go:47)
            if self.device != "cuda":
                return {"status": "skipped", "reason": "cpu device"}
            import time as _t
            from .profiler import GpuProfiler, NativeGpuProfiler
            try:
                gp = NativeGpuProfiler(self.profiles,
                                       process_name="deepflow-server")
            except Exception:   # roctracer unavailable -> kineto path
                gp = GpuProfiler(self.profiles,
                                 process_name="deepflow-server")
            with gp.capture():
                _t.sleep(min(seconds, 10.0))
            return {"status": "ok",
                    "rows": len(self.profiles.store.rows)}

        @self.app.post("/v1/datasources/")
        async def add_datasource(request: Request):
            body = await request.json()
            self.engine.add_datasource(body["name"],
                                       int(body["interval"]))
            return {"status": "ok"}

        @self.app.get("/v1/pcap/{flow_id}")
        def pcap_export(flow_id: int):
            from fastapi.responses import Response
            blob = self.pcap.export_pcap(flow_id)
            if blob is None:
                return Response(status_code=404)
            return Response(content=blob,
                            media_type="application/vnd.tcpdump.pcap")

        @self.app.get("/v1/export/otlp")
        def export_otlp(where: str = "", limit: int = 10000):
            from fastapi.responses import Response
            blob = self.exporter.export_where(where, limit)
            return Response(content=blob,
                            media_type="application/x-protobuf")

        # alert policies: DF-SQL rules evaluated on a timer -> alert_event
        from fastapi import Request
        from .control.alerting import AlertEvaluator, AlertPolicy
        self.alerts = AlertEvaluator(self.engine, self.events)

        @self.app.post("/v1/alert-policies/")
        async def add_alert_policy(request: Request):
            body = await request.json()
            self.alerts.add_policy(AlertPolicy(
                body["name"], body["sql"], body["column"],
                body.get("op", ">="), float(body["threshold"]),
                level=int(body.get("level", 2)),
                target_column=body.get("target_column")))
            return {"policies": len(self.alerts.policies)}

        @self.app.post("/v1/alert-policies/evaluate")
        def evaluate_alerts():
            return {"fired": self.alerts.evaluate_once()}

        # OTLP/HTTP collector endpoints (standard /v1/{traces,logs,metrics}
        # paths, protobuf bodies; reference accepts the same three
        # signals through its otel integration port)
        from fastapi import Request

        @self.app.post("/otlp/v1/traces")
        async def otlp_traces(request: Request):
            import numpy as np
            from .ingest.otel import otlp_to_l7_payload
            body = await request.body()
            try:
                l7_payload = otlp_to_l7_payload(body, compressed=False)
            except Exception:
                return {"error": "bad TracesData"}
            self._on_l7(framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG),
                        np.frombuffer(l7_payload, dtype=np.uint8))
            return {}

        @self.app.post("/otlp/v1/logs")
        async def otlp_logs(request: Request):
            from .ingest.otel import otlp_logs_to_rows
            body = await request.body()
            try:
                rows = otlp_logs_to_rows(body)
            except Exception:
                return {"error": "bad LogsData"}
            self.applogs.ingest_rows(rows)
            return {"accepted": len(rows)}

        @self.app.post("/otlp/v1/metrics")
        async def otlp_metrics(request: Request):
            from .ingest.otel import otlp_metrics_to_samples
            body = await request.body()
            try:
                samples = otlp_metrics_to_samples(body)
            except Exception:
                return {"error": "bad MetricsData"}
            n = self.prom.ingest_labeled_samples(samples)
            return {"accepted": n}

        # UDP debug command bus (deepflow-ctl ingester analog)
        from .utils.debug_bus import DebugBus
        from .utils.stats import default_registry as _dr
        self.debug_bus = DebugBus()
        self.debug_bus.register(
            "stats", lambda req: _dr().snapshot_all())
        self.debug_bus.register(
            "store", lambda req: {
                "l7_rows": self.l7.segments.n_rows,
                "l4_rows": self.l4.segments.n_rows,
                "l7_segments": len(self.l7.segments.segments),
                "l7_cold": len(getattr(self.l7.segments, "cold", [])),
                "dict_entries": self.l7.dict.n_entries(),
                "evicted_rows": self.l7.segments.evicted_rows,
            })
        self.debug_bus.register(
            "queues", lambda req: {
                "decode_queue": self.receiver._queue.qsize(),
                "drops": self.receiver.counter.snapshot(),
            })
        self.debug_bus.register(
            "agents", lambda req: {
                f"{aid}/{mt}": {"frames": st.frames, "bytes": st.bytes}
                for (aid, mt), st in self.receiver.status.items()})

        # one lock serializes GPU-store mutation (ingest) and scans
        # (queries): the engine shares it
        self._lock = threading.RLock()
        self.engine.lock = self._lock

        # checkpoint wiring (ckissu/metadb analog: restart-persistence
        # of the hot store + controller registry)
        self.checkpoint_dir: Optional[str] = None
        self._ckpt_thread: Optional[threading.Thread] = None

        @self.app.post("/v1/checkpoint/save")
        def checkpoint_save():
            if self.checkpoint_dir is None:
                return {"error": "no --checkpoint-dir configured"}
            return self.save_checkpoint(self.checkpoint_dir)

    # ------------------------------------------------------- checkpoint
    def save_checkpoint(self, ckpt_dir: str) -> dict:
        """Persist the L7 hot store + controller state atomically."""
        import os
        from .store import checkpoint as ck
        os.makedirs(ckpt_dir, exist_ok=True)
        with self._lock:
            ck.save_l7(self.l7, os.path.join(ckpt_dir, "l7.ckpt"))
            ck.save_l4(self.l4, os.path.join(ckpt_dir, "l4.ckpt"))
            import torch as _t
            tmp = os.path.join(ckpt_dir, "controller.ckpt.tmp")
            _t.save(self.controller.state_dict(), tmp)
            os.replace(tmp, os.path.join(ckpt_dir, "controller.ckpt"))
            tmp = os.path.join(ckpt_dir, "prom.ckpt.tmp")
            _t.save(self.prom.state_dict(), tmp)
            os.replace(tmp, os.path.join(ckpt_dir, "prom.ckpt"))
        return {"status": "ok", "rows": self.l7.segments.n_rows,
                "agents": len(self.controller.agents)}

    def load_checkpoint(self, ckpt_dir: str) -> dict:
        import os
        from .store import checkpoint as ck
        out = {"l7_rows": 0, "controller": False}
        p7 = os.path.join(ckpt_dir, "l7.ckpt")
        if os.path.exists(p7):
            with self._lock:
                out["l7_rows"] = ck.load_l7(self.l7, p7)
        p4 = os.path.join(ckpt_dir, "l4.ckpt")
        if os.path.exists(p4):
            with self._lock:
                out["l4_rows"] = ck.load_l4(self.l4, p4)
        pp = os.path.join(ckpt_dir, "prom.ckpt")
        if os.path.exists(pp):
            import torch as _t
            self.prom.load_state_dict(
                _t.load(pp, weights_only=False))
            out["prom"] = True
        pc = os.path.join(ckpt_dir, "controller.ckpt")
        if os.path.exists(pc):
            import torch as _t
            self.controller.load_state_dict(
                _t.load(pc, map_location="cpu", weights_only=False))
            out["controller"] = True
        return out

    def enable_checkpoints(self, ckpt_dir: str,
                           interval_s: float = 300.0,
                           load: bool = True) -> dict:
        """Startup load + periodic/background save loop."""
        self.checkpoint_dir = ckpt_dir
        restored = self.load_checkpoint(ckpt_dir) if load else {}
        if interval_s > 0:
            def loop():
                import time as _t
                while not self._stop_ckpt.wait(interval_s):
                    try:
                        self.save_checkpoint(ckpt_dir)
                    except Exception:
                        pass
            self._stop_ckpt = threading.Event()
            self._ckpt_thread = threading.Thread(target=loop, daemon=True)
            self._ckpt_thread.start()
        return restored

    # ------------------------------------------------------------------
    def org_context(self, org_id: int):
        """Lazily-created isolated (l7, l4, engine) triple for a
        non-default org."""
        if org_id in (0, self.default_org):
            return self
        ctx = self._org_ctx.get(org_id)
        if ctx is None:
            from types import SimpleNamespace
            from .ingest.l4_pipeline import L4IngestPipeline
            kg = KnowledgeGraphTable(device=self.device)
            l7 = L7IngestPipeline(device=self.device,
                                  segment_rows=self.l7.segments.segment_rows,
                                  kg=kg, dict_capacity=self.l7.dict.capacity,
                                  time_base_s=self.l7.time_base_s)
            l4 = L4IngestPipeline(device=self.device,
                                  segment_rows=self.l4.segments.segment_rows,
                                  kg=kg, time_base_s=self.l4.time_base_s)
            eng = QueryEngine(l7, device=self.device, l4_pipeline=l4)
            eng.lock = self._lock  # orgs share the ingest lock
            ctx = SimpleNamespace(l7=l7, l4=l4, engine=eng, kg=kg)
            self._org_ctx[org_id] = ctx
        return ctx

    def engine_for_org(self, org_id: int):
        return self.org_context(org_id).engine

    @staticmethod
    def _scan_records(payload):
        import ctypes as ct
        import numpy as np
        from .ops import native
        lib = native.cpu()
        max_n = max(len(payload) // 8, 16)
        offs = np.zeros(max_n, dtype=np.uint32)
        lens = np.zeros(max_n, dtype=np.uint32)
        n = int(lib.df_scan_offsets(payload.ctypes.data_as(ct.c_void_p),
                                    len(payload),
                                    offs.ctypes.data_as(ct.c_void_p),
                                    lens.ctypes.data_as(ct.c_void_p), max_n))
        return offs[:n].copy(), lens[:n].copy()

    def _on_pump_frame(self, view, meta):
        """PumpServer dispatch: frame metadata -> the registered
        per-type handler. L7 flow-log frames on a GPU server take the
        zero-copy coalescing feeder; everything else copies out of the
        ring and runs the normal handler."""
        msg_type, agent_id, org_id, team_id = meta
        if msg_type == framing.MSG_PROTOCOLLOG and \
                getattr(self, "_l7_feeder", None) is not None and \
                (org_id or 1) == self.default_org:
            return self._l7_feeder(view, meta)
        handler = self.receiver.handlers.get(msg_type)
        if handler is None:
            self.receiver.counter.add("unhandled_type")
            return None
        hdr = framing.FrameHeader(msg_type=msg_type, agent_id=agent_id,
                                  org_id=org_id or 1, team_id=team_id)
        handler(hdr, np.array(view, copy=True))
        self.receiver.counter.add("frames_in")
        return None

    def _on_l7(self, hdr, payload) -> None:
        offs, lens = self._scan_records(payload)
        pipe = self.org_context(hdr.org_id).l7
        with self._lock:
            pipe.ingest(payload, offs, lens)

    def _on_l4(self, hdr, payload) -> None:
        offs, lens = self._scan_records(payload)
        pipe = self.org_context(hdr.org_id).l4
        with self._lock:
            pipe.ingest(payload, offs, lens)

    def _on_otel(self, hdr, payload) -> None:
        """OTLP frames carry one zlib-compressed TracesData blob
        (reference decoder.go:235-266); convert to AppProtoLogsData and run
        the normal span pipeline. Conversion runs in C++ (OpenMP over
        spans, ops/csrc/otlp_conv.cpp — byte-identical to the Python
        twin) with the Python converter as fallback."""
        import numpy as np
        import zlib
        data = payload.tobytes()
        try:
            data = zlib.decompress(data)
        except zlib.error:
            pass  # uncompressed push
        from .ops import native
        lib = native.cpu()
        src = np.frombuffer(data, dtype=np.uint8)
        need = lib.df_otlp_to_l7(src.ctypes.data, len(src), None, 0)
        if need > 0:
            dst = np.zeros(int(need), dtype=np.uint8)
            lib.df_otlp_to_l7(src.ctypes.data, len(src),
                              dst.ctypes.data, int(need))
            self._on_l7(hdr, dst)
            return
        from .ingest.otel import otlp_to_l7_payload
        try:
            l7_payload = otlp_to_l7_payload(data, compressed=False)
        except Exception:
            self.receiver.counter.add("otel_decode_errors")
            return
        self._on_l7(hdr, np.frombuffer(l7_payload, dtype=np.uint8))

    def _on_third_party(self, hdr, payload, kind: str) -> None:
        import numpy as np
        from .ingest.thirdparty import third_party_frame_to_l7_payload
        try:
            l7_payload = third_party_frame_to_l7_payload(payload.tobytes(),
                                                         kind)
        except Exception:
            self.receiver.counter.add(f"{kind}_decode_errors")
            return
        self._on_l7(hdr, np.frombuffer(l7_payload, dtype=np.uint8))

    # ------------------------------------------------------------------
    def ingest_self_stats(self) -> int:
        """Self-telemetry loop: snapshot all Countables as dfstats records
        and feed them back through the receiver (reference: libs/stats
        shipping MESSAGE_TYPE_DFSTATS to its own ingester,
        SURVEY.md §5.H)."""
        from .utils.stats import default_registry
        payload = default_registry().encode_dfstats()
        hdr = framing.FrameHeader(msg_type=framing.MSG_DFSTATS)
        return 1 if self.receiver.handle_frame(
            framing.encode_frame(hdr, payload)) else 0

    def _on_dfstats(self, hdr, payload) -> None:
        from .wire import pb, metric
        for rec in framing.iter_records(payload.tobytes()):
            d = pb.decode(rec, metric.STATS)
            row = {"time": d.get("timestamp", 0),
                   "table": d.get("name", "")}
            for k, v in zip(d.get("tag_names", []), d.get("tag_values", [])):
                row[k] = v
            for k, v in zip(d.get("metrics_float_names", []),
                            d.get("metrics_float_values", [])):
                row[k] = v
            self.system_rows.append(row)

    def start(self) -> None:
        import gc
        # long-lived server state (dictionaries, segments, name maps)
        # stays out of gen2 GC scans — full collections over it showed
        # up as 60+ ms query-latency spikes under load
        gc.collect()
        gc.freeze()
        self.receiver.start()
        if self._use_pump:
            from .ingest.native_pump import PumpServer, GpuL7Feeder
            idle = None
            if self.device == "cuda":
                # GPU data plane: L7 frames bypass the per-frame python
                # dispatch — pinned views coalesce into ~1M-span device
                # ingests (the bench --path e2e fast path, as a product
                # component). The feeder's flushes serialize against
                # queries through the server-wide ingest lock.
                self._l7_feeder = GpuL7Feeder(self.l7,
                                              ingest_lock=self._lock)
                idle = self._l7_feeder.idle
            else:
                self._l7_feeder = None
            self.pump = PumpServer(self._on_pump_frame, port=self._pump_port,
                                   accept_type=-1,
                                   pin=self.device == "cuda",
                                   idle_handler=idle).start()
            self.pump_port = self.pump.port
        self.debug_bus.start()

    def stop(self) -> None:
        self.receiver.stop()
        if self.pump is not None:
            self.pump.stop()
        self.debug_bus.stop()
        if self._ckpt_thread is not None:
            self._stop_ckpt.set()
        if self.checkpoint_dir is not None:
            try:
                self.save_checkpoint(self.checkpoint_dir)
            except Exception:
                pass

    def serve_http(self, host: str = "127.0.0.1", port: int = 20416) -> None:
        import uvicorn
        uvicorn.run(self.app, host=host, port=port, log_level="warning")


def main() -> None:
    import argparse
    # native GPU profiler registration must precede HIP init
    try:
        from .profiler.native_profiler import ensure_early
        ensure_early()
    except Exception:
        pass
    import torch
    ap = argparse.ArgumentParser(description="deepflow-amd all-in-one server")
    ap.add_argument("--device", default=None)
    ap.add_argument("--tcp-port", type=int, default=20033)
    ap.add_argument("--http-port", type=int, default=20416)
    ap.add_argument("--checkpoint-dir", default=None,
                    help="load on start, save every --checkpoint-interval "
                         "seconds and on shutdown")
    ap.add_argument("--checkpoint-interval", type=float, default=300.0)
    args = ap.parse_args()
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    srv = DeepflowServer(device=device, tcp_port=args.tcp_port)
    if args.checkpoint_dir:
        restored = srv.enable_checkpoints(args.checkpoint_dir,
                                          args.checkpoint_interval)
        print(f"checkpoint restore: {restored}")
    srv.start()
    try:
        srv.serve_http(port=args.http_port)
    finally:
        srv.stop()


if __name__ == "__main__":
    main()
