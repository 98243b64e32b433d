"""L7 (span) ingest pipeline: payload -> decode -> KG join -> SmartEncoding
intern -> string-pool gather -> metric rollup -> columnar segment.

GPU mode launches the HIP kernels (K1/K2/K3/K4/K5) asynchronously on the
current torch stream with ONE host sync per batch (the combined
attr/string pool sizing readback); the dictionary harvest defers behind
a CUDA event (steady state = no sync) and the naive-bytes accounting
accumulates on-device. CPU mode runs the reference ops; both modes
produce identical logical contents (tests/test_pipeline_cpu.py,
tests/test_gpu_pipeline.py).

Reference call-stack being replaced: receiver -> flow_log Decoder.Run ->
L7FlowLog.Fill (KnowledgeGraph join) -> FlowTag -> CKWriter
(server/ingester/flow_log/decoder/decoder.go:151-232).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, Optional

import numpy as np
import torch

from ..store import l7_schema as S
from ..store.segment import SegmentSet, L7Segment
from ..store.dictionary import TagDictionary
from ..store.kg import KnowledgeGraphTable
from ..store.metrics import L7_TABLES, RollupFamily
from ..utils.stats import Counter


_SCALAR_DICT_REF_ROWS = [src for (_, src, _) in S.DID_COLS]
_SCALAR_DICT_DOMAINS = [dom for (_, _, dom) in S.DID_COLS]


@dataclass
class PipelineStats:
    spans_in: int = 0
    batches: int = 0
    dict_new: int = 0
    pool_bytes: int = 0
    # bytes the batch's tag/trace strings would cost stored verbatim
    # (the ClickHouse String-column baseline SmartEncoding is measured
    # against; numeric columns excluded on both sides)
    naive_str_bytes: int = 0


class L7IngestPipeline:
    def __init__(self, device: str = "cpu", segment_rows: int = 1 << 22,
                 time_base_s: int = 1_700_000_000,
                 kg: Optional[KnowledgeGraphTable] = None,
                 dictionary: Optional[TagDictionary] = None,
                 dict_capacity: int = 1 << 22,
                 window_bytes: Optional[int] = None,
                 counter: Optional[Counter] = None,
                 defer_harvest: bool = False):
        # defer_harvest=True moves the per-batch dictionary harvest off
        # the hot path (event-gated, resolves next batch / at query time
        # via engine.sync_stats). Callers that read pipe.dict directly
        # after ingest, or recycle the payload buffer immediately (native
        # pump ring), need the synchronous default
        self.device = device
        self.segments = SegmentSet(segment_rows, device,
                                   max_bytes=window_bytes)
        self.kg = kg or KnowledgeGraphTable(device=device)
        self.dict = dictionary or TagDictionary(dict_capacity, device=device)
        # flow_metrics application table family (application{,_map}.{1s,1m})
        self.rollups = RollupFamily(L7_TABLES, time_base_s, device=device)
        self.metrics = self.rollups.get("application.1s")
        self.time_base_s = time_base_s
        self.stats = PipelineStats()
        self.counter = counter or Counter("ingester.l7")
        dev = torch.device(device)
        self._ref_rows_scalar = torch.tensor(_SCALAR_DICT_REF_ROWS,
                                             dtype=torch.int16, device=dev)
        self._dom_scalar = torch.tensor(_SCALAR_DICT_DOMAINS,
                                        dtype=torch.uint8, device=dev)
        self._ref_rows_attr = torch.arange(2 * S.MAX_ATTRS, dtype=torch.int16,
                                           device=dev)
        self._dom_attr = torch.tensor(
            [S.DICT_DOM_ATTR_NAME] * S.MAX_ATTRS +
            [S.DICT_DOM_ATTR_VALUE] * S.MAX_ATTRS,
            dtype=torch.uint8, device=dev)
        self._pool_cols = torch.tensor(S.POOL_COLS, dtype=torch.uint8,
                                       device=dev)
        # per-batch scratch (string refs are transient; only pooled refs
        # persist into segments)
        self._scratch_str = None
        self._scratch_attr = None
        # async bookkeeping (GPU mode): naive-bytes accounting accumulates
        # on-device; dictionary harvest is deferred behind an event so the
        # steady state (no new strings) costs zero host syncs
        self.defer_harvest = defer_harvest
        if device != "cpu":
            self._naive_dev = torch.zeros(1, dtype=torch.int64, device=dev)
            self._h_cnt = torch.zeros(1, dtype=torch.int32,
                                      pin_memory=torch.cuda.is_available())
            self._h_ev = torch.cuda.Event() if torch.cuda.is_available() \
                else None
            self._h_payload = None

    # ------------------------------------------------------------------
    def ingest(self, payload: np.ndarray, offs: np.ndarray,
               lens: np.ndarray) -> int:
        """Ingest one pre-segmented batch. payload/offs/lens are host numpy
        arrays (payload uint8, offs/lens uint32). Returns rows ingested."""
        n = len(offs)
        if n == 0:
            return 0
        seg = self.segments.tail(n)
        base = seg.n_rows
        if self.device == "cpu":
            self._ingest_cpu(payload, offs, lens, seg, base, n)
        else:
            self._ingest_gpu(payload, offs, lens, seg, base, n)
        seg.n_rows += n
        self.stats.spans_in += n
        self.stats.batches += 1
        self.counter.add("spans_in", n)
        return n

    # ------------------------------------------------------------------
    def ingest_device(self, payload_t: torch.Tensor, offs_t: torch.Tensor,
                      lens_t: torch.Tensor, payload_host) -> int:
        """GPU fast path: batch tensors already on-device (benchmarks /
        receivers prefetch H2D on a side stream). payload_host is the same
        bytes on the host (numpy), used only for dictionary harvest."""
        n = offs_t.numel()
        if n == 0:
            return 0
        seg = self.segments.tail(n)
        base = seg.n_rows
        self._run_gpu(payload_t, offs_t, lens_t, payload_host, seg, base, n)
        seg.n_rows += n
        self.stats.spans_in += n
        self.stats.batches += 1
        self.counter.add("spans_in", n)
        return n

    def _ingest_gpu(self, payload, offs, lens, seg: L7Segment, base: int,
                    n: int) -> None:
        dev = torch.device(self.device)
        payload_t = torch.from_numpy(payload).to(dev, non_blocking=True)
        offs_t = torch.from_numpy(offs.view(np.int32)).to(dev, non_blocking=True)
        lens_t = torch.from_numpy(lens.view(np.int32)).to(dev, non_blocking=True)
        self._run_gpu(payload_t, offs_t, lens_t, payload, seg, base, n)

    def _scratch(self, n: int, dev):
        if self._scratch_str is None or self._scratch_str.shape[1] < n:
            self._scratch_str = torch.zeros((S.N_STR, n), dtype=torch.int64,
                                            device=dev)
            self._scratch_attr = torch.zeros((2 * S.MAX_ATTRS, n),
                                             dtype=torch.int64, device=dev)
        else:
            self._scratch_str.zero_()
            self._scratch_attr.zero_()
        return self._scratch_str, self._scratch_attr

    def _run_gpu(self, payload_t, offs_t, lens_t, payload_host,
                 seg: L7Segment, base: int, n: int) -> None:
        from ..ops import gpu_ops
        dev = payload_t.device
        # resolve the PREVIOUS batch's deferred harvest first: its emit
        # rows reference that batch's payload and must drain before this
        # batch's intern kernels can append new rows
        self._harvest_pending()
        sstr, sattr = self._scratch(n, dev)
        gpu_ops.decode_l7(payload_t, offs_t, lens_t, seg, base, sstr, sattr)
        gpu_ops.intern_many(payload_t, sstr, self._ref_rows_scalar,
                            self._dom_scalar, 0, n, self.dict.tkeys,
                            self.dict.emit, self.dict.emit_ctr, seg.did, base)
        # sizing: attr-pool cumsum (from decode's attr_cnt) and string-pool
        # cumsum (from decode's sstr) both launch async, then ONE host sync
        # reads both totals — the only sync in the batch
        cnts = seg.attr_cnt[base:base + n].to(torch.int64) * 2
        acum = torch.cumsum(cnts, 0)
        row_len = torch.zeros(n, dtype=torch.int32, device=dev)
        gpu_ops.pool_lens(sstr, self._pool_cols, n, row_len)
        cum = torch.cumsum(row_len.to(torch.int64), 0)
        totals = torch.stack((acum[-1], cum[-1])).cpu()
        attr_total, total = int(totals[0]), int(totals[1])
        seg.ensure_attr_pool(attr_total)
        starts = (acum - cnts + seg.attr_pool_len).to(torch.int32)
        seg.attr_start[base:base + n] = starts
        gpu_ops.intern_attrs(payload_t, seg, base, n, self.dict.tkeys,
                             self.dict.emit, self.dict.emit_ctr, sattr,
                             starts)
        seg.attr_pool_len += attr_total
        row_start = cum - row_len.to(torch.int64)
        seg.ensure_pool(total)
        gpu_ops.pool_gather(payload_t, seg, self._pool_cols, base, n,
                            row_start, seg.pool, seg.pool_len, sstr)
        self.rollups.update(seg, base, n)
        # naive-string accounting accumulates on-device (read via
        # sync_stats); the old per-batch int(...) forced two extra syncs
        self._naive_dev += (sstr[:, :n] & 0xFFFF).sum() + \
            (sattr[:, :n] & 0xFFFF).sum()
        self._harvest_defer(payload_host)
        seg.pool_len += total
        self.stats.pool_bytes += total

    # -------------------------------------------------- deferred harvest
    def _harvest_defer(self, payload_host) -> None:
        """Queue an async emit-counter readback; the actual harvest (if
        any strings were new) happens at the start of the next batch or
        at sync_stats()."""
        if self._h_ev is None or not self.defer_harvest:
            self.stats.dict_new += self.dict.harvest(payload_host)
            return
        self._h_cnt.copy_(self.dict.emit_ctr, non_blocking=True)
        self._h_ev.record()
        self._h_payload = payload_host

    def _harvest_pending(self) -> None:
        if self._h_payload is None:
            return
        self._h_ev.synchronize()  # waits only for the tiny D2H copy
        if int(self._h_cnt[0]) > 0:
            self.stats.dict_new += self.dict.harvest(self._h_payload)
        self._h_payload = None

    def sync_stats(self) -> PipelineStats:
        """Flush deferred bookkeeping into `stats` (call after a sync)."""
        if self.device != "cpu":
            self._harvest_pending()
            self.stats.naive_str_bytes = int(self._naive_dev.item())
        return self.stats

    # ------------------------------------------------------------------
    def _ingest_cpu(self, payload, offs, lens, seg: L7Segment, base: int,
                    n: int) -> None:
        from ..ops import ref
        pb = payload.tobytes()
        sstr, sattr = self._scratch(n, torch.device("cpu"))
        ref.decode_l7_ref(pb, offs, lens, seg, base, sstr, sattr)
        new = ref.intern_ref(pb, sstr, _SCALAR_DICT_REF_ROWS,
                             _SCALAR_DICT_DOMAINS, 0, n, self.dict.tkeys,
                             seg.did, base, dictionary=self.dict)
        new += ref.intern_attrs_pool_ref(pb, sattr, seg, base, n,
                                         self.dict.tkeys,
                                         dictionary=self.dict)
        row_len = ref.pool_lens_ref(sstr, S.POOL_COLS, n)
        cum = torch.cumsum(row_len.to(torch.int64), 0)
        total = int(cum[-1].item()) if n else 0
        row_start = cum - row_len.to(torch.int64)
        seg.ensure_pool(total)
        ref.pool_gather_ref(pb, seg, S.POOL_COLS, base, n, row_start,
                            seg.pool_len, sstr)
        self.rollups.update(seg, base, n)
        naive = int((sstr[:, :n] & 0xFFFF).sum()) + \
            int((sattr[:, :n] & 0xFFFF).sum())
        self.stats.naive_str_bytes += naive
        seg.pool_len += total
        self.stats.dict_new += len(new)
        self.stats.pool_bytes += total

    # ------------------------------------------------------------------
    def ingest_frame_payload(self, payload: bytes) -> int:
        """Convenience: scan offsets (native) then ingest."""
        from ..ops import native
        import ctypes as ct
        arr = np.frombuffer(payload, dtype=np.uint8)
        max_n = max(len(payload) // 8, 16)
        offs = np.zeros(max_n, dtype=np.uint32)
        lens = np.zeros(max_n, dtype=np.uint32)
        lib = native.cpu()
        n = lib.df_scan_offsets(arr.ctypes.data_as(ct.c_void_p), len(payload),
                                offs.ctypes.data_as(ct.c_void_p),
                                lens.ctypes.data_as(ct.c_void_p), max_n)
        n = int(n)
        return self.ingest(arr, offs[:n].copy(), lens[:n].copy())
