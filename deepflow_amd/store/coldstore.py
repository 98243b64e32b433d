"""Cold-segment compression: bit-packed HBM-resident columns.

When the hot window exceeds its watermark, full segments are demoted in
place of being dropped: every fixed-width integer column is re-encoded as
(base, bits, packed words) — a frame-of-reference + bit-pack codec run by
the k_pack_bits kernel on the GPU (word-centric, no atomics). Constant
columns collapse to zero payload. Queries materialize a compressed segment
back into a recycled scratch segment on demand (k_unpack_bits).

Reference analog: ClickHouse column codecs (T64/DoubleDelta) on cold
parts + ckmonitor's forced partition drops; here the demotion keeps the
data queryable inside the 288 GB HBM window instead of deleting it.
"""
from __future__ import annotations

import math
from typing import Dict, List, Optional, Tuple

import torch


def _bits_for(rng: int) -> int:
    if rng <= 0:
        return 0
    return max(1, math.ceil(math.log2(rng + 1)))


def _pack_words_cpu(vals: torch.Tensor, bits: int) -> torch.Tensor:
    """CPU reference pack: same layout as k_pack_bits (little-endian bit
    order within the u32 word stream)."""
    import numpy as np
    v = vals.numpy().astype(np.uint64)
    n = len(v)
    out_words = (n * bits + 31) // 32
    out = np.zeros(out_words, dtype=np.uint64)  # u64 scratch, fold later
    bitpos = np.arange(n, dtype=np.uint64) * np.uint64(bits)
    w = (bitpos >> np.uint64(5)).astype(np.int64)
    off = bitpos & np.uint64(31)
    lo = (v << off) & np.uint64(0xFFFFFFFF)
    hi = (v >> (np.uint64(32) - off)) if bits else v * 0
    # values with off+bits<=32 contribute nothing to the next word
    np.bitwise_or.at(out, w, lo)
    spill = (off + bits) > 32
    np.bitwise_or.at(out, w[spill] + 1, hi[spill])
    return torch.from_numpy(out.astype(np.uint32).view(np.int32))


def _unpack_words_cpu(packed: torch.Tensor, n: int, bits: int) -> torch.Tensor:
    import numpy as np
    pw = packed.numpy().view(np.uint32).astype(np.uint64)
    pw = np.append(pw, np.uint64(0))
    bitpos = np.arange(n, dtype=np.uint64) * np.uint64(bits)
    w = (bitpos >> np.uint64(5)).astype(np.int64)
    off = bitpos & np.uint64(31)
    both = pw[w] | (pw[w + 1] << np.uint64(32))
    mask = np.uint64((1 << bits) - 1) if bits < 64 else np.uint64(-1)
    vals = (both >> off) & mask
    return torch.from_numpy(vals.astype(np.uint32).view(np.int32))


def pack_stream(vals_u32: torch.Tensor, base: int, bits: int,
                stream: int = 0) -> torch.Tensor:
    """Pack a contiguous int32 tensor (values already >= base,
    range < 2^bits) into ceil(n*bits/32) packed words."""
    n = vals_u32.numel()
    out_words = (n * bits + 31) // 32
    if vals_u32.device.type == "cpu":
        shifted = (vals_u32.to(torch.int64) -
                   base) & 0xFFFFFFFF
        return _pack_words_cpu(shifted.to(torch.int64), bits)
    from ..ops import native
    lib = native.gpu()
    out = torch.zeros(out_words, dtype=torch.int32, device=vals_u32.device)
    rc = lib.df_pack_bits(vals_u32.data_ptr(), n, base & 0xFFFFFFFF, bits,
                          out.data_ptr(), out_words, stream)
    if rc != 0:
        raise RuntimeError(f"df_pack_bits failed: {rc}")
    return out


def unpack_stream(packed: torch.Tensor, n: int, base: int, bits: int,
                  out: Optional[torch.Tensor] = None,
                  stream: int = 0) -> torch.Tensor:
    if out is None:
        out = torch.empty(n, dtype=torch.int32, device=packed.device)
    if packed.device.type == "cpu":
        vals = _unpack_words_cpu(packed, n, bits)
        out.copy_((vals.to(torch.int64) + base).to(torch.int32))
        return out
    from ..ops import native
    lib = native.gpu()
    rc = lib.df_unpack_bits(packed.data_ptr(), n, base & 0xFFFFFFFF, bits,
                            out.data_ptr(), stream)
    if rc != 0:
        raise RuntimeError(f"df_unpack_bits failed: {rc}")
    return out


class PackedColumn:
    """One compressed column row: frame-of-reference bit pack, or raw
    fallback when the range does not fit 32 bits."""

    __slots__ = ("base", "bits", "data", "n", "raw")

    def __init__(self, base: int, bits: int, data: Optional[torch.Tensor],
                 n: int, raw: Optional[torch.Tensor] = None):
        self.base = base
        self.bits = bits
        self.data = data
        self.n = n
        self.raw = raw

    def nbytes(self) -> int:
        if self.raw is not None:
            return self.raw.numel() * self.raw.element_size()
        return 0 if self.data is None else self.data.numel() * 4


def _compress_i32_matrix(mat: torch.Tensor, n: int,
                         stream: int = 0) -> List[PackedColumn]:
    """Pack each row of an (ncols, capacity) int32/int64 matrix over its
    first n entries."""
    cols = []
    for c in range(mat.shape[0]):
        col = mat[c, :n]
        col64 = col.to(torch.int64)
        if mat.dtype == torch.int32:
            col64 = col64 & 0xFFFFFFFF  # stored unsigned
        mn = int(col64.min()) if n else 0
        mx = int(col64.max()) if n else 0
        rng = mx - mn
        if rng >= (1 << 32):  # raw fallback (u64 column, wide range)
            cols.append(PackedColumn(0, 64, None, n, raw=col.clone()))
            continue
        bits = _bits_for(rng)
        if bits == 0:
            cols.append(PackedColumn(mn, 0, None, n))
            continue
        shifted = (col64 - mn).to(torch.int32)
        cols.append(PackedColumn(mn, bits,
                                 pack_stream(shifted, 0, bits, stream), n))
    return cols


def _restore_i32_matrix(cols: List[PackedColumn], mat: torch.Tensor,
                        n: int, stream: int = 0,
                        only: Optional[set] = None) -> None:
    for c, pc in enumerate(cols):
        if only is not None and c not in only:
            continue
        dst = mat[c, :n]
        if pc.raw is not None:
            dst.copy_(pc.raw)
        elif pc.bits == 0:
            dst.fill_(pc.base if mat.dtype != torch.int32
                      else _as_i32(pc.base))
        else:
            vals = unpack_stream(pc.data, n, 0, pc.bits, stream=stream)
            restored = (vals.to(torch.int64) & 0xFFFFFFFF) + pc.base
            if mat.dtype == torch.int32:
                dst.copy_(_wrap_i32(restored))
            else:
                dst.copy_(restored)


def _as_i32(v: int) -> int:
    v &= 0xFFFFFFFF
    return v - (1 << 32) if v >= (1 << 31) else v


def _wrap_i32(t: torch.Tensor) -> torch.Tensor:
    t = t & 0xFFFFFFFF
    return torch.where(t >= (1 << 31), t - (1 << 32), t).to(torch.int32)


class CompressedL7Segment:
    """Bit-packed demoted segment; queryable after materialize().
    Handles both L7 and L4 segment layouts (L4 has no dict-id or attr
    blocks — those compress to empty column lists)."""

    is_compressed = True

    def __init__(self, seg, stream: int = 0):
        n = seg.n_rows
        self.n_rows = n
        self.capacity = seg.capacity
        self.device = seg.device
        self.layout_version = getattr(seg, "layout_version", 0)
        # u64 columns: 64-bit frame-of-reference (delta almost always
        # fits 32 bits: times share the segment window, rrt/lens are small)
        self.u64_cols = _compress_i32_matrix(seg.u64, n, stream)
        self.u32_cols = _compress_i32_matrix(seg.u32, n, stream)
        self.did_cols = _compress_i32_matrix(seg.did, n, stream) \
            if hasattr(seg, "did") else []
        self.rowref_col = _compress_i32_matrix(seg.str_rowref.view(1, -1),
                                               n, stream)
        # small/raw blocks (u8 is already 1 B/row; pools are variable)
        self.u8 = seg.u8[:, :n].clone()
        self.str_lens = seg.str_lens[:, :n].clone()
        if hasattr(seg, "attr_start"):
            self.attr_start = seg.attr_start[:n].clone()
            self.attr_cnt = seg.attr_cnt[:n].clone()
            self.attr_pool = seg.attr_pool[: seg.attr_pool_len].clone()
            self.attr_pool_len = seg.attr_pool_len
        else:
            self.attr_start = None
            self.attr_cnt = None
            self.attr_pool = None
            self.attr_pool_len = 0
        self.pool = seg.pool[: seg.pool_len].clone()
        self.pool_len = seg.pool_len
        # time bounds of u64 col 0 (start_time, absolute ns): queries
        # with a time predicate skip cold segments entirely outside it
        if n:
            t = seg.u64[0, :n]
            self.time_min = int(t.min().item()) & ((1 << 64) - 1)
            self.time_max = int(t.max().item()) & ((1 << 64) - 1)
        else:
            self.time_min = 0
            self.time_max = 0

    def compressed_bytes(self) -> int:
        total = 0
        for group in (self.u64_cols, self.u32_cols, self.did_cols,
                      self.rowref_col):
            total += sum(pc.nbytes() for pc in group)
        for t in (self.u8, self.str_lens, self.attr_start, self.attr_cnt,
                  self.attr_pool, self.pool):
            if t is not None:
                total += t.numel() * t.element_size()
        return total

    def materialize(self, seg, stream: int = 0, needed=None):
        """Decompress into a recycled L7Segment (capacity >= n_rows).

        `needed` (optional): {family: set(col_idx) | None} from the query
        plan — only the referenced columns are unpacked (a scan touches a
        handful of the ~60 columns; unpacking just those makes cold scans
        ~10x cheaper). None = everything (select-row fetch paths).
        NB: untouched scratch columns hold stale/zero data, which is safe
        because the kernel's access pattern is exactly the plan's."""
        from ..query import spec as Q
        n = self.n_rows
        assert seg.capacity >= n

        def want(fam):
            if needed is None:
                return None
            return needed.get(fam, set())

        _restore_i32_matrix(self.u64_cols, seg.u64, n, stream,
                            only=want(Q.SRC_U64))
        _restore_i32_matrix(self.u32_cols, seg.u32, n, stream,
                            only=want(Q.SRC_U32))
        if self.did_cols:
            _restore_i32_matrix(self.did_cols, seg.did, n, stream,
                                only=want(Q.SRC_DID))
        u8_only = want(Q.SRC_U8)
        if u8_only is None:
            seg.u8[:, :n] = self.u8
        else:
            for c in u8_only:
                seg.u8[c, :n] = self.u8[c]
        strings_needed = needed is None or Q.SRC_STR_HASH in needed
        attrs_needed = needed is None or Q.SRC_ATTR_MATCH in needed or             Q.SRC_ATTR_VAL in needed
        if strings_needed:
            _restore_i32_matrix(self.rowref_col,
                                seg.str_rowref.view(1, -1), n, stream)
            seg.str_lens[:, :n] = self.str_lens
            seg.ensure_pool(self.pool_len)
            seg.pool[: self.pool_len] = self.pool
            seg.pool_len = self.pool_len
        if attrs_needed and self.attr_start is not None:
            seg.attr_start[:n] = self.attr_start
            seg.attr_cnt[:n] = self.attr_cnt
            seg.ensure_attr_pool(self.attr_pool_len)
            seg.attr_pool[: self.attr_pool_len] = self.attr_pool
            seg.attr_pool_len = self.attr_pool_len
        seg.n_rows = n
        return seg
