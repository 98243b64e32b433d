"""HBM-resident columnar segment for l7_flow_log.

One `L7Segment` is a fixed-capacity SoA block; the decode/join/intern kernels
write rows in place (no row->column transpose pass: the decoder IS the
transpose, writing straight into column slices). A `SegmentSet` chains
segments into the shard's hot window, sized for 288 GB HBM per GPU.

Device-agnostic: tensors live on 'cuda' on a GPU box (written by HIP
kernels) or 'cpu' in tests (written by ops/ref.py reference ops).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from . import l7_schema as S


class L7Segment:
    def __init__(self, capacity: int, device: str = "cpu",
                 pool_capacity: Optional[int] = None):
        self.capacity = capacity
        self.device = device
        dev = torch.device(device)
        z = lambda shape, dt: torch.zeros(shape, dtype=dt, device=dev)
        self.u64 = z((S.N_U64, capacity), torch.int64)
        self.u32 = z((S.N_U32, capacity), torch.int32)
        self.u8 = z((S.N_U8, capacity), torch.uint8)
        self.str_rowref = z((capacity,), torch.int64)
        self.str_lens = z((S.N_POOL, capacity), torch.int16)
        self.did = torch.full((S.N_DID, capacity), -1, dtype=torch.int32, device=dev)
        # KG ids are NOT stored per row: the query engine joins them from
        # the (epc, ip) key against the GPU-resident platform table
        # (SmartEncoding — saves 2*N_KG*4 = 96 B/span of HBM)
        self.attr_start = z((capacity,), torch.int32)
        self.attr_pool = torch.full((capacity * 8,), -1, dtype=torch.int32,
                                    device=dev)
        self.attr_pool_len = 0
        self.attr_cnt = z((capacity,), torch.uint8)
        self.pool = z((pool_capacity or capacity * 96,), torch.uint8)
        self.pool_len = 0
        self.n_rows = 0
        self.layout_version = S.LAYOUT_VERSION

    def ensure_attr_pool(self, extra: int) -> None:
        need = self.attr_pool_len + extra
        if need > self.attr_pool.numel():
            new_cap = max(need, self.attr_pool.numel() * 2)
            newp = torch.full((new_cap,), -1, dtype=torch.int32,
                              device=self.attr_pool.device)
            newp[: self.attr_pool_len] = self.attr_pool[: self.attr_pool_len]
            self.attr_pool = newp

    def free_rows(self) -> int:
        return self.capacity - self.n_rows

    def ensure_pool(self, extra: int) -> None:
        need = self.pool_len + extra
        if need > self.pool.numel():
            new_cap = max(need, self.pool.numel() * 2)
            new_pool = torch.zeros(new_cap, dtype=torch.uint8,
                                   device=self.pool.device)
            new_pool[: self.pool_len] = self.pool[: self.pool_len]
            self.pool = new_pool

    def stored_bytes_per_row(self) -> float:
        """Bytes/span actually resident (SmartEncoding accounting).

        Counts the fixed-width blocks that survive past ingest (u64/u32/u8,
        dict ids, KG ids, attr ids + count) plus the variable pool. The
        transient strref/attr_ref blocks are working state for the batch in
        flight, not storage, but we count strref's pooled refs (needed to
        address the pool).
        """
        if self.n_rows == 0:
            return 0.0
        fixed = (S.N_U64 * 8 + S.N_U32 * 4 + S.N_U8 * 1 + S.N_DID * 4 +
                 4 + 1 + 8 + S.N_POOL * 2)
        return fixed + (self.pool_len + 4 * self.attr_pool_len) / self.n_rows


class L4Segment:
    """Columnar segment for l4_flow_log (layout: l4_schema / l4_layout.h)."""

    def __init__(self, capacity: int, device: str = "cpu",
                 pool_capacity: Optional[int] = None):
        from . import l4_schema as L4
        self.capacity = capacity
        self.device = device
        dev = torch.device(device)
        z = lambda shape, dt: torch.zeros(shape, dtype=dt, device=dev)
        self.u64 = z((L4.N_U64, capacity), torch.int64)
        self.u32 = z((L4.N_U32, capacity), torch.int32)
        self.u8 = z((L4.N_U8, capacity), torch.uint8)
        self.str_rowref = z((capacity,), torch.int64)
        self.str_lens = z((L4.N_STR, capacity), torch.int16)
        self.pool = z((pool_capacity or capacity * 24,), torch.uint8)
        self.pool_len = 0
        self.n_rows = 0

    def free_rows(self) -> int:
        return self.capacity - self.n_rows

    def ensure_pool(self, extra: int) -> None:
        need = self.pool_len + extra
        if need > self.pool.numel():
            new_cap = max(need, self.pool.numel() * 2)
            new_pool = torch.zeros(new_cap, dtype=torch.uint8,
                                   device=self.pool.device)
            new_pool[: self.pool_len] = self.pool[: self.pool_len]
            self.pool = new_pool

    def stored_bytes_per_row(self) -> float:
        from . import l4_schema as L4
        if self.n_rows == 0:
            return 0.0
        fixed = (L4.N_U64 * 8 + L4.N_U32 * 4 + L4.N_U8 +
                 8 + L4.N_STR * 2)
        return fixed + self.pool_len / self.n_rows


def _hot_overlaps(seg, lo: int, hi: int) -> bool:
    """Hot-segment time pruning: per-segment [min,max] of u64 col 0
    (start_time), cached until the row count changes. The tail segment
    recomputes as it grows — one tiny device reduction per query."""
    n = seg.n_rows
    if n == 0:
        return False
    cache = getattr(seg, "_tb_cache", None)
    if cache is None or cache[0] != n:
        t = seg.u64[0, :n]
        cache = (n, int(t.min().item()) & ((1 << 64) - 1),
                 int(t.max().item()) & ((1 << 64) - 1))
        seg._tb_cache = cache
    return cache[2] >= lo and cache[1] <= hi


class SegmentSet:
    """The shard-local hot window: ordered list of segments."""

    def __init__(self, segment_rows: int, device: str = "cpu", cls=None,
                 max_bytes: Optional[int] = None):
        self.segment_rows = segment_rows
        self.device = device
        self.cls = cls or L7Segment
        self.segments: List = []
        # HBM hot-window watermark (ckmonitor analog: reference
        # ckmonitor/monitor.go:339-432 force-drops oldest partitions)
        self.max_bytes = max_bytes
        self.evicted_rows = 0
        self.evicted_segments = 0
        self._free: List = []

    @staticmethod
    def seg_alloc_bytes(seg) -> int:
        total = 0
        for name in ("u64", "u32", "u8", "str_rowref", "str_lens", "did",
                     "attr_start", "attr_pool", "attr_cnt", "pool"):
            t = getattr(seg, name, None)
            if t is not None:
                total += t.numel() * t.element_size()
        return total

    def total_alloc_bytes(self) -> int:
        # scratch segments (cold-query materialization) are live HBM too
        return sum(self.seg_alloc_bytes(s) for s in self.segments) + \
            sum(self.seg_alloc_bytes(s) for s in getattr(self, "_scratch", []))

    def enforce_watermark(self) -> int:
        """Drop oldest full segments while over the byte watermark; dropped
        segments go to a free-list and are zero-reset on reuse (steady-state
        segment rolls are memsets, not allocations). Returns segments
        evicted this call."""
        if self.max_bytes is None:
            return 0
        dropped = 0

        def over():
            return self.total_alloc_bytes() + sum(
                self.seg_alloc_bytes(s) for s in self._free) + sum(
                c.compressed_bytes() for c in getattr(self, "cold", [])) > \
                self.max_bytes

        # oldest cold segments go first (they were demoted earliest)
        while getattr(self, "cold", []) and over():
            c = self.cold.pop(0)
            self.evicted_rows += c.n_rows
            self.evicted_segments += 1
            dropped += 1
        while len(self.segments) > 1 and over():
            seg = self.segments.pop(0)
            self.evicted_rows += seg.n_rows
            self.evicted_segments += 1
            self._free.append(seg)
            dropped += 1
        return dropped

    @staticmethod
    def reset_segment(seg) -> None:
        for name in ("u64", "u32", "u8", "str_rowref", "str_lens",
                     "attr_start"):
            t = getattr(seg, name, None)
            if t is not None:
                t.zero_()
        for name in ("did", "attr_pool"):
            t = getattr(seg, name, None)
            if t is not None:
                t.fill_(-1)
        t = getattr(seg, "attr_cnt", None)
        if t is not None:
            t.zero_()
        seg.pool_len = 0
        if hasattr(seg, "attr_pool_len"):
            seg.attr_pool_len = 0
        seg.n_rows = 0

    def reserve(self, n_segments: int) -> None:
        """Provision the hot window up front: pre-allocate n_segments into
        the free list so steady-state segment rolls never hit the device
        allocator (size-for-288GB-HBM startup provisioning)."""
        while len(self.segments) + len(self._free) < n_segments:
            self._free.append(self.cls(self.segment_rows, self.device))

    def tail(self, min_free: int):
        if not self.segments or self.segments[-1].free_rows() < min_free:
            if self._free:
                seg = self._free.pop(0)
                self.reset_segment(seg)
                self.segments.append(seg)
            else:
                self.segments.append(self.cls(self.segment_rows, self.device))
            # two-tier policy: over the hot watermark -> demote (compress)
            # oldest hot segments; over max_bytes -> drop oldest cold
            hot_max = getattr(self, "hot_max_bytes", None)
            if hot_max is not None:
                while (len(self.segments) > 1 and
                       self.total_alloc_bytes() > hot_max):
                    if not self.demote_oldest():
                        break
            self.enforce_watermark()
        return self.segments[-1]

    @property
    def n_rows(self) -> int:
        return sum(s.n_rows for s in self.segments) + \
            sum(c.n_rows for c in getattr(self, "cold", []))

    def total_stored_bytes(self) -> int:
        total = 0
        for s in self.segments:
            total += int(s.stored_bytes_per_row() * s.n_rows)
        for c in getattr(self, "cold", []):
            total += c.compressed_bytes()
        return total

    # -------------------------------------------------- cold demotion
    def demote_oldest(self, stream: int = 0) -> bool:
        """Compress the oldest full hot segment into the cold tier
        (store/coldstore.py bit-pack codec); its buffers go to the
        free-list. Returns False if nothing is demotable."""
        if len(self.segments) <= 1:
            return False
        from .coldstore import CompressedL7Segment
        if not hasattr(self, "cold"):
            self.cold = []
        seg = self.segments.pop(0)
        self.cold.append(CompressedL7Segment(seg, stream))
        self._free.append(seg)
        return True

    def scan_list(self, stream: int = 0, needed=None,
                  time_range=None) -> List:
        """All queryable segments: cold ones materialized into recycled
        scratch segments. `needed` restricts decompression to the columns
        a query plan touches; `time_range` = (lo_ns, hi_ns) prunes cold
        segments whose recorded [time_min, time_max] lies entirely
        outside the predicate (the reference's partition pruning). Call
        release_scratch() when the query is done — the scratch buffers
        go back to the free-list so they are reused by ingest and
        counted by the watermark."""
        cold = getattr(self, "cold", [])
        hot = self.segments
        if time_range is not None:
            lo, hi = time_range
            cold = [c for c in cold
                    if getattr(c, "time_max", None) is None or
                    (c.time_max >= lo and c.time_min <= hi)]
            hot = [g for g in hot
                   if _hot_overlaps(g, lo, hi)]
        if not cold:
            return hot
        out = []
        if not hasattr(self, "_scratch"):
            self._scratch = []
        while len(self._scratch) < len(cold):
            if self._free:
                seg = self._free.pop(0)
            else:
                seg = self.cls(self.segment_rows, self.device)
            self._scratch.append(seg)
        for c, scratch in zip(cold, self._scratch):
            self.reset_segment(scratch)
            out.append(c.materialize(scratch, stream, needed=needed))
        return out + hot

    def release_scratch(self) -> None:
        """Return cold-query scratch segments to the free-list (bounds the
        scratch pool to the lifetime of one query instead of forever)."""
        scratch = getattr(self, "_scratch", None)
        if scratch:
            self._free.extend(scratch)
            self._scratch = []
