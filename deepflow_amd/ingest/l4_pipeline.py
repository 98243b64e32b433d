"""L4 (TaggedFlow) ingest pipeline: decode -> KG join -> pool -> network.1s.

Reference counterpart: flow_log L4 Logger + L4FlowLog.Fill
(server/ingester/flow_log/log_data/l4_flow_log.go) + network metric rollups.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import numpy as np
import torch

from ..store import l4_schema as L4
from ..store import l7_schema as S
from ..store.segment import SegmentSet, L4Segment
from ..store.kg import KnowledgeGraphTable
from ..store.metrics import L4_TABLES, RollupFamily
from ..utils.stats import Counter


@dataclass
class L4Stats:
    flows_in: int = 0
    batches: int = 0


class L4IngestPipeline:
    def __init__(self, device: str = "cpu", segment_rows: int = 1 << 22,
                 time_base_s: int = 1_700_000_000,
                 kg: Optional[KnowledgeGraphTable] = None,
                 counter: Optional[Counter] = None):
        self.device = device
        self.segments = SegmentSet(segment_rows, device, cls=L4Segment)
        self.kg = kg or KnowledgeGraphTable(device=device)
        # flow_metrics network table family (network{,_map}.{1s,1m} +
        # traffic_policy.1m)
        self.rollups = RollupFamily(L4_TABLES, time_base_s, device=device)
        self.metrics = self.rollups.get("network.1s")
        self.time_base_s = time_base_s
        self.stats = L4Stats()
        self.counter = counter or Counter("ingester.l4")

    def ingest(self, payload: np.ndarray, offs: np.ndarray,
               lens: np.ndarray) -> int:
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
        self.stats.flows_in += n
        self.stats.batches += 1
        self.counter.add("flows_in", n)
        return n

    def _scratch(self, n: int, dev):
        if getattr(self, "_scratch_str", None) is None or \
                self._scratch_str.shape[1] < n:
            self._scratch_str = torch.zeros((L4.N_STR, n), dtype=torch.int64,
                                            device=dev)
        else:
            self._scratch_str.zero_()
        return self._scratch_str

    def _ingest_gpu(self, payload, offs, lens, seg, base, n) -> None:
        from ..ops import gpu_ops
        dev = torch.device(self.device)
        payload_t = torch.from_numpy(payload).to(dev, non_blocking=True)
        offs_t = torch.from_numpy(offs.view(np.int32)).to(dev, non_blocking=True)
        lens_t = torch.from_numpy(lens.view(np.int32)).to(dev, non_blocking=True)
        sstr = self._scratch(n, dev)
        gpu_ops.decode_l4(payload_t, offs_t, lens_t, seg, base, sstr)
        # pool the string columns (request_domain + ip6 pair)
        pool_cols = torch.arange(L4.N_STR, dtype=torch.uint8, device=dev)
        row_len = torch.zeros(n, dtype=torch.int32, device=dev)
        gpu_ops.pool_lens(sstr, pool_cols, n, row_len)
        cum = torch.cumsum(row_len.to(torch.int64), 0)
        total = int(cum[-1].item())
        seg.ensure_pool(total)
        if total:
            starts = cum - row_len.to(torch.int64)
            gpu_ops.pool_gather(payload_t, seg, pool_cols, base, n, starts,
                                seg.pool, seg.pool_len, sstr)
        self.rollups.update(seg, base, n)
        seg.pool_len += total

    def _ingest_cpu(self, payload, offs, lens, seg, base, n) -> None:
        from ..ops import ref_l4, ref
        pbytes = payload.tobytes()
        sstr = self._scratch(n, torch.device("cpu"))
        ref_l4.decode_l4_ref(pbytes, offs, lens, seg, base, sstr)
        pool_cols = list(range(L4.N_STR))
        row_len = ref.pool_lens_ref(sstr, pool_cols, n)
        cum = torch.cumsum(row_len.to(torch.int64), 0)
        total = int(cum[-1].item()) if n else 0
        seg.ensure_pool(total)
        starts = cum - row_len.to(torch.int64)
        ref.pool_gather_ref(pbytes, seg, pool_cols, base, n, starts,
                            seg.pool_len, sstr)
        self.rollups.update(seg, base, n)
        seg.pool_len += total

    def ingest_frame_payload(self, payload: bytes) -> int:
        from ..ops import native
        import ctypes as ct
        arr = np.frombuffer(payload, dtype=np.uint8)
        max_n = max(len(payload) // 8, 16)
        offs = np.zeros(max_n, dtype=np.uint32)
        lens = np.zeros(max_n, dtype=np.uint32)
        lib = native.cpu()
        n = int(lib.df_scan_offsets(arr.ctypes.data_as(ct.c_void_p),
                                    len(payload),
                                    offs.ctypes.data_as(ct.c_void_p),
                                    lens.ctypes.data_as(ct.c_void_p), max_n))
        return self.ingest(arr, offs[:n].copy(), lens[:n].copy())
