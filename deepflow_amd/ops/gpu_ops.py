"""Tensor-level wrappers over the HIP C API (libdfgpu.so).

Every function takes torch tensors already resident on the GPU, launches
asynchronously on the current torch CUDA stream, and raises on HIP errors.
There is deliberately NO eager/PyTorch fallback here: on a GPU box these are
the only code paths (see ops/ref.py for the CPU reference used by tests and
by the device='cpu' pipeline).
"""
from __future__ import annotations

import torch

from . import native


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def decode_l7(payload: torch.Tensor, offs: torch.Tensor, lens: torch.Tensor,
              seg, base_row: int, scratch_str: torch.Tensor,
              scratch_attr: torch.Tensor) -> None:
    n = offs.numel()
    lib = native.gpu()
    native.check(lib.df_decode_l7(
        payload.data_ptr(), offs.data_ptr(), lens.data_ptr(), n,
        seg.u64.data_ptr(), seg.u32.data_ptr(), seg.u8.data_ptr(),
        scratch_str.data_ptr(), scratch_attr.data_ptr(),
        seg.attr_cnt.data_ptr(),
        seg.capacity, base_row, scratch_str.shape[1], _stream()),
        "df_decode_l7")


def decode_l4(payload: torch.Tensor, offs: torch.Tensor, lens: torch.Tensor,
              seg, base_row: int, scratch_str: torch.Tensor) -> None:
    n = offs.numel()
    lib = native.gpu()
    native.check(lib.df_decode_l4(
        payload.data_ptr(), offs.data_ptr(), lens.data_ptr(), n,
        seg.u64.data_ptr(), seg.u32.data_ptr(), seg.u8.data_ptr(),
        scratch_str.data_ptr(), seg.capacity, base_row,
        scratch_str.shape[1], _stream()), "df_decode_l4")


def rollup_family(seg, base_row: int, n: int, time_base_s: int, tables,
                  stream: int = 0) -> None:
    """ONE fused launch updates every table of a family (k_rollup_l4/l7
    iterate the per-table key specs over a single pass of the row)."""
    import numpy as np
    lib = native.gpu()
    src = tables[0].td.source
    fn = lib.df_rollup_l4 if src == "l4" else lib.df_rollup_l7
    specs = b"".join(t.spec_bytes() for t in tables)
    ptrs = np.array([x for t in tables
                     for x in (t.tkeys.data_ptr(), t.traw.data_ptr(),
                               t.tvals.data_ptr(), t.drops.data_ptr())],
                    dtype=np.uint64)
    caps = np.array([t.capacity for t in tables], dtype=np.uint32)
    native.check(fn(
        seg.u64.data_ptr(), seg.u32.data_ptr(), seg.u8.data_ptr(),
        seg.capacity, base_row, n, time_base_s, specs, len(tables),
        ptrs.ctypes.data, caps.ctypes.data,
        stream or _stream()), "df_rollup_family")


def rollup_insert(kws: torch.Tensor, vals: torch.Tensor, ops: torch.Tensor,
                  table, stream: int = 0) -> None:
    lib = native.gpu()
    native.check(lib.df_rollup_insert(
        kws.data_ptr(), vals.data_ptr(), ops.data_ptr(),
        kws.shape[0], kws.shape[1], vals.shape[1],
        table.tkeys.data_ptr(), table.traw.data_ptr(),
        table.tvals.data_ptr(), table.capacity, table.drops.data_ptr(),
        stream or _stream()), "df_rollup_insert")


def sort_u64(data: torch.Tensor) -> None:
    """In-place rocPRIM radix sort of a device int64 tensor (values
    treated as u64)."""
    import ctypes
    lib = native.gpu()
    n = data.numel()
    if n == 0:
        return
    alt = torch.empty_like(data)
    nbytes = ctypes.c_uint64(0)
    native.check(lib.df_sort_u64(data.data_ptr(), alt.data_ptr(), n, None,
                                 ctypes.byref(nbytes), _stream()),
                 "df_sort_u64(size)")
    temp = torch.empty(int(nbytes.value), dtype=torch.uint8,
                       device=data.device)
    native.check(lib.df_sort_u64(data.data_ptr(), alt.data_ptr(), n,
                                 temp.data_ptr(), ctypes.byref(nbytes),
                                 _stream()), "df_sort_u64")


def gather_records(payload: torch.Tensor, offs: torch.Tensor,
                   lens: torch.Tensor, sel: torch.Tensor,
                   dst_off: torch.Tensor, out: torch.Tensor) -> None:
    """Pack selected records contiguously (shard-routing all-to-all)."""
    lib = native.gpu()
    native.check(lib.df_gather_records(
        payload.data_ptr(), offs.data_ptr(), lens.data_ptr(),
        sel.data_ptr(), dst_off.data_ptr(), sel.numel(), out.data_ptr(),
        _stream()), "df_gather_records")


def kg_build(keys: torch.Tensor, vals: torch.Tensor, tkeys: torch.Tensor,
             tvals: torch.Tensor) -> None:
    lib = native.gpu()
    native.check(lib.df_kg_build(
        keys.data_ptr(), vals.data_ptr(), keys.numel(),
        tkeys.data_ptr(), tvals.data_ptr(), tkeys.numel(), _stream()),
        "df_kg_build")


def intern_many(payload: torch.Tensor, refs: torch.Tensor,
                ref_rows: torch.Tensor, domains: torch.Tensor,
                ref_base_row: int, n: int,
                tkeys: torch.Tensor, emit: torch.Tensor,
                emit_ctr: torch.Tensor, out_ids: torch.Tensor,
                out_base_row: int) -> None:
    C = domains.numel()
    lib = native.gpu()
    native.check(lib.df_intern_many(
        payload.data_ptr(), refs.data_ptr(), ref_rows.data_ptr(),
        domains.data_ptr(), C, n, refs.shape[1], ref_base_row,
        tkeys.data_ptr(), tkeys.numel(),
        emit.data_ptr(), emit_ctr.data_ptr(), emit.shape[0],
        out_ids.data_ptr(), out_ids.shape[1], out_base_row, _stream()),
        "df_intern_many")


def intern_attrs(payload: torch.Tensor, seg, base_row: int, n: int,
                 tkeys: torch.Tensor, emit: torch.Tensor,
                 emit_ctr: torch.Tensor, scratch_attr: torch.Tensor,
                 attr_start: torch.Tensor) -> None:
    lib = native.gpu()
    native.check(lib.df_intern_attrs(
        payload.data_ptr(), scratch_attr.data_ptr(), seg.attr_cnt.data_ptr(),
        n, seg.capacity, base_row, scratch_attr.shape[1],
        tkeys.data_ptr(), tkeys.numel(),
        emit.data_ptr(), emit_ctr.data_ptr(), emit.shape[0],
        attr_start.data_ptr(), seg.attr_pool.data_ptr(), _stream()),
        "df_intern_attrs")


def pool_lens(scratch_str: torch.Tensor, pool_cols: torch.Tensor, n: int,
              row_len: torch.Tensor) -> None:
    lib = native.gpu()
    native.check(lib.df_pool_lens(
        scratch_str.data_ptr(), pool_cols.data_ptr(), pool_cols.numel(), n,
        scratch_str.shape[1], 0, row_len.data_ptr(), _stream()),
        "df_pool_lens")


def pool_gather(payload: torch.Tensor, seg, pool_cols: torch.Tensor,
                base_row: int, n: int, row_start: torch.Tensor,
                pool: torch.Tensor, pool_base: int,
                scratch_str: torch.Tensor) -> None:
    lib = native.gpu()
    native.check(lib.df_pool_gather(
        payload.data_ptr(), scratch_str.data_ptr(), pool_cols.data_ptr(),
        pool_cols.numel(), n, scratch_str.shape[1], 0, row_start.data_ptr(),
        pool.data_ptr(), pool_base,
        seg.str_rowref.data_ptr(), seg.str_lens.data_ptr(),
        seg.capacity, base_row, _stream()),
        "df_pool_gather")



def _opt_ptr(seg, attr: str) -> int:
    t = getattr(seg, attr, None)
    return t.data_ptr() if t is not None else 0


def query_agg(seg, spec_bytes: bytes, base_row: int, n: int,
              gkeys: torch.Tensor, graw: torch.Tensor,
              gvals: torch.Tensor, kg=None) -> None:
    import ctypes
    lib = native.gpu()
    buf = ctypes.create_string_buffer(spec_bytes, len(spec_bytes))
    ktk, ktv, kcap = _kg_args(kg)
    native.check(lib.df_query_agg(
        seg.u64.data_ptr(), seg.u32.data_ptr(), seg.u8.data_ptr(),
        _opt_ptr(seg, "did"), ktk, ktv, kcap, _opt_ptr(seg, "attr_pool"),
        _opt_ptr(seg, "attr_start"), _opt_ptr(seg, "attr_cnt"),
        seg.str_rowref.data_ptr(), seg.str_lens.data_ptr(),
        seg.pool.data_ptr(), seg.capacity, seg.n_rows,
        ctypes.addressof(buf), n, base_row,
        gkeys.data_ptr(), graw.data_ptr(), gvals.data_ptr(), gkeys.numel(),
        _stream()), "df_query_agg")


def qpart_agg(seg, spec_bytes: bytes, base_row: int, n: int,
              counts: torch.Tensor, cursors: torch.Tensor,
              row_scratch: torch.Tensor,
              gkeys: torch.Tensor, graw: torch.Tensor,
              gvals: torch.Tensor, kg=None) -> None:
    """Radix-partitioned group-by (high key cardinality): bucket rows by
    8 hash bits, aggregate one bucket per workgroup set — each block's
    LDS table then holds every key it sees (no per-row global atomics).
    counts/cursors: u32[256] (counts zeroed); row_scratch:\n    u64[n * (n_keys+n_aggs)] payload staging."""
    import ctypes
    lib = native.gpu()
    buf = ctypes.create_string_buffer(spec_bytes, len(spec_bytes))
    ktk, ktv, kcap = _kg_args(kg)
    native.check(lib.df_qpart_agg(
        seg.u64.data_ptr(), seg.u32.data_ptr(), seg.u8.data_ptr(),
        _opt_ptr(seg, "did"), ktk, ktv, kcap, _opt_ptr(seg, "attr_pool"),
        _opt_ptr(seg, "attr_start"), _opt_ptr(seg, "attr_cnt"),
        seg.str_rowref.data_ptr(), seg.str_lens.data_ptr(),
        seg.pool.data_ptr(), seg.capacity, seg.n_rows,
        ctypes.addressof(buf), n, base_row,
        counts.data_ptr(), cursors.data_ptr(), row_scratch.data_ptr(),
        gkeys.data_ptr(), graw.data_ptr(), gvals.data_ptr(),
        gkeys.numel(), _stream()), "df_qpart_agg")


def _kg_args(kg):
    """KnowledgeGraph table pointers for the query-time join."""
    if kg is None:
        return 0, 0, 0
    return kg.tkeys.data_ptr(), kg.tvals.data_ptr(), kg.tkeys.numel()


def query_select(seg, spec_bytes: bytes, base_row: int, n: int,
                 out_rows: torch.Tensor, out_ctr: torch.Tensor,
                 kg=None) -> None:
    import ctypes
    lib = native.gpu()
    buf = ctypes.create_string_buffer(spec_bytes, len(spec_bytes))
    ktk, ktv, kcap = _kg_args(kg)
    native.check(lib.df_query_select(
        seg.u64.data_ptr(), seg.u32.data_ptr(), seg.u8.data_ptr(),
        _opt_ptr(seg, "did"), ktk, ktv, kcap, _opt_ptr(seg, "attr_pool"),
        _opt_ptr(seg, "attr_start"), _opt_ptr(seg, "attr_cnt"),
        seg.str_rowref.data_ptr(), seg.str_lens.data_ptr(),
        seg.pool.data_ptr(), seg.capacity, seg.n_rows,
        ctypes.addressof(buf), n, base_row,
        out_rows.data_ptr(), out_ctr.data_ptr(), out_rows.numel(), _stream()),
        "df_query_select")
