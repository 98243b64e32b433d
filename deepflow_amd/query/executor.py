"""Query executor: runs a Plan over the shard's segments.

GPU: k_query_agg / k_query_select (one launch per segment, shared group
table). CPU: vectorized numpy oracle with identical semantics — used by
tests as the numerics reference for the GPU kernels and by device='cpu'
deployments.
"""
from __future__ import annotations

from typing import Dict, List

import numpy as np
import torch


from .spec import (Plan, SRC_U64, SRC_U32, SRC_U8, SRC_DID, SRC_KG,
                   SRC_ATTR_VAL, SRC_TIME_BUCKET, SRC_CONST0, SRC_STR_HASH,
                   SRC_ATTR_MATCH, SRC_TRACE128, STR_FILTER_SEED,
                   OP_EQ, OP_NE, OP_LT, OP_LE, OP_GT, OP_GE, OP_BETWEEN,
                   AGGOP_COUNT, AGGOP_SUM, AGGOP_MIN, AGGOP_MAX,
                   QMAX_KEYS, QMAX_AGGS)

U64MAX = (1 << 64) - 1
GROUP_CAP = 1 << 20


def _src_np(seg, family: int, idx: int, bucket: int, time_base_s: int,
            n: int, kg=None) -> np.ndarray:
    """Vectorized src_value over rows [0, n) as uint64 (CPU oracle)."""
    def u(t):
        return t[:, :n][idx].numpy()
    if family == SRC_U64:
        return seg.u64[idx, :n].numpy().view(np.uint64)
    if family == SRC_U32:
        return seg.u32[idx, :n].numpy().view(np.uint32).astype(np.uint64)
    if family == SRC_U8:
        return seg.u8[idx, :n].numpy().astype(np.uint64)
    if family == SRC_DID:
        return seg.did[idx, :n].numpy().view(np.uint32).astype(np.uint64)
    if family == SRC_KG:
        # query-time KnowledgeGraph join from the row's (epc, ip) key
        # (KG ids are not materialized per row — SmartEncoding)
        from ..store import l7_schema as S7
        side, j = idx // S7.N_KG, idx % S7.N_KG
        epc = seg.u32[3 + side, :n].numpy().view(np.uint32)
        ip = seg.u32[1 + side, :n].numpy().view(np.uint32)
        out = np.zeros(n, dtype=np.uint64)
        if kg is not None:
            host = kg.host
            cache = {}
            for i in range(n):
                key = (int(epc[i]), int(ip[i]))
                v = cache.get(key)
                if v is None:
                    info = host.get(key)
                    v = info.as_list()[j] if info is not None else 0
                    cache[key] = v
                out[i] = v
        return out
    if family == SRC_ATTR_VAL:
        starts = seg.attr_start[:n].numpy()
        cnts = seg.attr_cnt[:n].numpy()
        pool = seg.attr_pool.numpy()
        out = np.full(n, 0xFFFFFFFF, dtype=np.uint64)
        for i in range(n):
            if idx < cnts[i]:
                out[i] = np.uint32(pool[starts[i] + cnts[i] + idx])
        return out
    if family == SRC_TRACE128:
        from ..ops.ref import mix64
        from ..store import l7_schema as S7
        if idx == 0:
            hi = seg.u64[S7.U64_COLS.index("trace_id_hi"), :n].numpy().view(np.uint64)
            lo = seg.u64[S7.U64_COLS.index("trace_id_lo"), :n].numpy().view(np.uint64)
            fb = _src_np(seg, SRC_STR_HASH, S7.POOL_POS["trace_id"], 0,
                         time_base_s, n)
            out = np.empty(n, dtype=np.uint64)
            for i in range(n):
                out[i] = (mix64(int(hi[i])) ^ int(lo[i])) \
                    if (int(hi[i]) | int(lo[i])) else int(fb[i])
            return out
        sv = seg.u64[S7.U64_COLS.index("span_id_b"), :n].numpy().view(np.uint64)
        fb = _src_np(seg, SRC_STR_HASH, S7.POOL_POS["span_id"], 0,
                     time_base_s, n)
        return np.where(sv != 0, sv, fb)
    if family == SRC_TIME_BUCKET:
        t_s = seg.u64[0, :n].numpy().view(np.uint64) // np.uint64(10**9)
        rel = np.maximum(t_s.astype(np.int64) - time_base_s, 0).astype(np.uint64)
        if bucket:
            rel = (rel // np.uint64(bucket)) * np.uint64(bucket)
        return rel
    if family == SRC_STR_HASH:
        from ..store.dictionary import str_hash_py
        rowref = seg.str_rowref[:n].numpy().view(np.uint64)
        lens = seg.str_lens[:, :n].numpy().astype(np.uint16)
        pool = seg.pool.numpy().tobytes()
        out = np.zeros(n, dtype=np.uint64)
        for i in range(n):
            ln = int(lens[idx, i])
            if ln:
                off = (int(rowref[i]) >> 16) + int(lens[:idx, i].sum())
                out[i] = str_hash_py(pool[off:off + ln], STR_FILTER_SEED)
        return out
    return np.zeros(n, dtype=np.uint64)


def _term_mask(seg, t, plan: Plan, n: int, kg=None) -> np.ndarray:
    if t.family == SRC_ATTR_MATCH:
        starts = seg.attr_start[:n].numpy()
        cnts = seg.attr_cnt[:n].numpy()
        pool = seg.attr_pool.numpy()
        out = np.zeros(n, dtype=bool)
        for i in range(n):
            c = int(cnts[i])
            s0 = int(starts[i])
            for a in range(c):
                if (int(pool[s0 + a]) & 0xFFFFFFFF) == t.v0 and \
                        (int(pool[s0 + c + a]) & 0xFFFFFFFF) == t.v1:
                    out[i] = True
                    break
        if t.op == OP_NE:
            out = ~out
        return out
    v = _src_np(seg, t.family, t.idx, 0, plan.time_base_s, n, kg=kg)
    v0 = np.uint64(t.v0 & U64MAX)
    v1 = np.uint64(t.v1 & U64MAX)
    if t.op == OP_EQ:
        return v == v0
    if t.op == OP_NE:
        return v != v0
    if t.op == OP_LT:
        return v < v0
    if t.op == OP_LE:
        return v <= v0
    if t.op == OP_GT:
        return v > v0
    if t.op == OP_GE:
        return v >= v0
    if t.op == OP_BETWEEN:
        return (v >= v0) & (v <= v1)
    return np.ones(n, dtype=bool)


def _mask_np(seg, plan: Plan, n: int, kg=None) -> np.ndarray:
    mask = np.ones(n, dtype=bool)
    groups = {}
    for t in plan.terms:
        tm = _term_mask(seg, t, plan, n, kg=kg)
        g = getattr(t, "group", 0)
        if g == 0:
            mask &= tm
        else:
            groups[g] = groups.get(g, np.zeros(n, dtype=bool)) | tm
    for gm in groups.values():
        mask &= gm
    return mask


def execute_agg_cpu(plan: Plan, segments, kg=None) -> List[Dict]:
    groups: Dict[tuple, List[int]] = {}
    for seg in segments:
        n = seg.n_rows
        if n == 0:
            continue
        mask = _mask_np(seg, plan, n, kg=kg)
        if not mask.any():
            continue
        keys = [_src_np(seg, k.family, k.idx, k.bucket, plan.time_base_s, n,
                        kg=kg)[mask]
                for k in plan.keys]
        aggvals = []
        for a in plan.aggs:
            if a.op == AGGOP_COUNT:
                aggvals.append(np.ones(int(mask.sum()), dtype=np.uint64))
            else:
                aggvals.append(_src_np(seg, a.family, a.idx, 0,
                                       plan.time_base_s, n, kg=kg)[mask])
        nk = len(keys)
        rows = int(mask.sum())
        key_tup = np.empty((rows, nk), dtype=np.uint64)
        for j, kv in enumerate(keys):
            key_tup[:, j] = kv
        for r in range(rows):
            kt = tuple(int(x) for x in key_tup[r])
            acc = groups.get(kt)
            if acc is None:
                acc = []
                for a in plan.aggs:
                    acc.append(0 if a.op in (AGGOP_COUNT, AGGOP_SUM)
                               else (U64MAX if a.op == AGGOP_MIN else 0))
                groups[kt] = acc
            for ai, a in enumerate(plan.aggs):
                v = int(aggvals[ai][r])
                if a.op in (AGGOP_COUNT, AGGOP_SUM):
                    acc[ai] += v
                elif a.op == AGGOP_MIN:
                    acc[ai] = min(acc[ai], v)
                else:
                    acc[ai] = max(acc[ai], v)
    return [{"key": list(k), "agg": list(v)} for k, v in groups.items()]


# observed group-cardinality per key-set: repeated dashboard queries on a
# high-cardinality GROUP BY switch to the radix-partitioned kernel after
# the first run discovers the cardinality
_CARDINALITY_CACHE: Dict[tuple, int] = {}
_QPART_MIN_ROWS = 1 << 22
_QPART_MIN_GROUPS = 1024
# partition scratch persists across queries: a fresh ~134 MB hipMalloc
# costs tens of ms and showed up as 20-80 ms query-latency spikes
_QPART_SCRATCH: Dict[str, tuple] = {}


def _qpart_scratch(dev, need_elems: int):
    key = str(dev)
    sc = _QPART_SCRATCH.get(key)
    if sc is None or sc[2].numel() < need_elems:
        sc = (torch.empty(256, dtype=torch.int32, device=dev),
              torch.empty(256, dtype=torch.int32, device=dev),
              torch.empty(max(need_elems, 1 << 20), dtype=torch.int64,
                          device=dev))
        _QPART_SCRATCH[key] = sc
    return sc


def execute_agg_gpu(plan: Plan, segments, device="cuda", kg=None) -> List[Dict]:
    from ..ops import gpu_ops
    dev = torch.device(device)
    gkeys = torch.zeros(GROUP_CAP, dtype=torch.int64, device=dev)
    graw = torch.zeros((GROUP_CAP, QMAX_KEYS), dtype=torch.int64, device=dev)
    gvals = torch.zeros((GROUP_CAP, QMAX_AGGS), dtype=torch.int64, device=dev)
    for ai, a in enumerate(plan.aggs):
        if a.op == AGGOP_MIN:
            gvals[:, ai] = -1  # 0xFFFF.. as int64
    spec = plan.to_bytes()
    key_sig = tuple((k.family, k.idx, k.bucket) for k in plan.keys)
    known_card = _CARDINALITY_CACHE.get(key_sig, 0)
    for seg in segments:
        if seg.n_rows == 0:
            continue
        if known_card >= _QPART_MIN_GROUPS and \
                seg.n_rows >= _QPART_MIN_ROWS and plan.keys:
            w = len(plan.keys) + len(plan.aggs)
            scratch = _qpart_scratch(dev, seg.n_rows * w)
            scratch[0].zero_()
            gpu_ops.qpart_agg(seg, spec, 0, seg.n_rows, scratch[0],
                              scratch[1], scratch[2], gkeys, graw, gvals,
                              kg=kg)
        else:
            gpu_ops.query_agg(seg, spec, 0, seg.n_rows, gkeys, graw,
                              gvals, kg=kg)
    torch.cuda.synchronize()
    mask = gkeys != 0
    raw = graw[mask].cpu().numpy().view(np.uint64)
    vals = gvals[mask].cpu().numpy().view(np.uint64)
    nk, na = len(plan.keys), len(plan.aggs)
    # .tolist() converts at C speed; per-element numpy indexing was the
    # dominant wall cost on multi-thousand-group results
    raw_l = raw[:, :nk].tolist()
    vals_l = vals[:, :na].tolist()
    out = [{"key": k, "agg": v} for k, v in zip(raw_l, vals_l)]
    if plan.keys:
        _CARDINALITY_CACHE[key_sig] = len(out)
    return out


def execute_select_cpu(plan: Plan, segments, limit: int,
                       kg=None) -> List[int]:
    """Returns (segment_idx, row) pairs encoded as global row ids."""
    out = []
    for si, seg in enumerate(segments):
        n = seg.n_rows
        if n == 0:
            continue
        mask = _mask_np(seg, plan, n, kg=kg)
        rows = np.nonzero(mask)[0]
        for r in rows:
            out.append((si, int(r)))
            if len(out) >= limit:
                return out
    return out


def execute_select_gpu(plan: Plan, segments, limit: int,
                       device="cuda", kg=None) -> List[int]:
    from ..ops import gpu_ops
    dev = torch.device(device)
    out = []
    spec = plan.to_bytes()
    for si, seg in enumerate(segments):
        if seg.n_rows == 0:
            continue
        cap = min(limit * 4 + 1024, 1 << 22)
        out_rows = torch.zeros(cap, dtype=torch.int64, device=dev)
        out_ctr = torch.zeros(1, dtype=torch.int32, device=dev)
        gpu_ops.query_select(seg, spec, 0, seg.n_rows, out_rows, out_ctr,
                             kg=kg)
        torch.cuda.synchronize()
        cnt = int(out_ctr.item())
        if cnt > cap:
            # the atomic-emit kernel counted every match but only the
            # first `cap` landed (arrival order is nondeterministic);
            # rerun with an exact-size buffer so LIMIT/ORDER BY sees
            # the deterministic full match set
            out_rows = torch.zeros(cnt, dtype=torch.int64, device=dev)
            out_ctr.zero_()
            gpu_ops.query_select(seg, spec, 0, seg.n_rows, out_rows,
                                 out_ctr, kg=kg)
            torch.cuda.synchronize()
            cnt = min(int(out_ctr.item()), cnt)
        rows = sorted(out_rows[:cnt].cpu().tolist())
        for r in rows:
            out.append((si, int(r)))
            if len(out) >= limit:
                return out
    return out


def execute(plan: Plan, segments, device: str = "cpu", kg=None):
    if plan.impossible:
        return []
    if plan.select_rows:
        limit = plan.limit or 100
        if device == "cpu":
            return execute_select_cpu(plan, segments, limit, kg=kg)
        return execute_select_gpu(plan, segments, limit, device, kg=kg)
    if device == "cpu":
        return execute_agg_cpu(plan, segments, kg=kg)
    return execute_agg_gpu(plan, segments, device, kg=kg)


# ---------------------------------------------------------------------
# grouped value gather for Percentile/Apdex: one pass over the store
# (filter -> per-row group assignment -> two-key stable sort), replacing
# the per-group re-scan and its 256-group cap. Reference analog:
# ClickHouse quantile() aggregate states.
# ---------------------------------------------------------------------
_M32 = 0xFFFFFFFF


def _kg_lookup_torch(kg, seg, rows_t, idx):
    """Vectorized open-addressing probe of the KG table for the selected
    rows (query-time join, device-resident)."""
    from ..store import l7_schema as S7
    side, j = idx // S7.N_KG, idx % S7.N_KG
    epc = seg.u32[3 + side].index_select(0, rows_t).to(torch.int64) & _M32
    ip = seg.u32[1 + side].index_select(0, rows_t).to(torch.int64) & _M32
    key = (epc << 32) | ip
    if kg is None:
        return torch.zeros_like(key)
    tk, tv = kg.tkeys, kg.tvals
    if tk.device != key.device:
        key = key.to(tk.device)
    mask = tk.numel() - 1
    # exact u64 mix64 via numpy (torch int64 shifts are arithmetic)
    import numpy as np
    kn = key.cpu().numpy().view(np.uint64)
    zn = kn.copy()
    zn = (zn ^ (zn >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
    zn = (zn ^ (zn >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
    zn = zn ^ (zn >> np.uint64(31))
    slot = torch.from_numpy((zn & np.uint64(mask)).astype(np.int64)).to(
        tk.device)
    out = torch.zeros(key.numel(), dtype=torch.int64, device=tk.device)
    live = torch.ones(key.numel(), dtype=torch.bool, device=tk.device)
    for _ in range(64):  # bounded probe chain
        tkv = tk.index_select(0, slot)
        hit = live & (tkv == key)
        if hit.any():
            out[hit] = tv.view(-1)[slot[hit] * tv.shape[1] + j].to(
                torch.int64) & _M32
        live = live & (tkv != key) & (tkv != 0)
        if not bool(live.any()):
            break
        slot = torch.where(live, (slot + 1) & mask, slot)
    return out.to(rows_t.device) if out.device != rows_t.device else out


def _col_torch(seg, family, idx, bucket, time_base_s, rows_t, kg=None):
    if family == SRC_U64:
        return seg.u64[idx].index_select(0, rows_t)
    if family == SRC_U32:
        return seg.u32[idx].index_select(0, rows_t).to(torch.int64) & _M32
    if family == SRC_U8:
        return seg.u8[idx].index_select(0, rows_t).to(torch.int64)
    if family == SRC_DID:
        return seg.did[idx].index_select(0, rows_t).to(torch.int64) & _M32
    if family == SRC_KG:
        return _kg_lookup_torch(kg, seg, rows_t, idx)
    if family == SRC_TIME_BUCKET:
        t_s = seg.u64[0].index_select(0, rows_t) // 10**9
        rel = torch.clamp(t_s - time_base_s, min=0)
        if bucket:
            rel = (rel // bucket) * bucket
        return rel
    raise ValueError(f"unsupported tensor family {family}")


def _matching_rows_t(plan: Plan, seg, device, kg=None):
    n = seg.n_rows
    if device == "cpu" or seg.u64.device.type == "cpu":
        mask = _mask_np(seg, plan, n, kg=kg)
        return torch.from_numpy(np.nonzero(mask)[0].copy())
    from ..ops import gpu_ops
    dev = seg.u64.device
    cap = n
    out_rows = torch.zeros(cap, dtype=torch.int64, device=dev)
    out_ctr = torch.zeros(1, dtype=torch.int32, device=dev)
    spec = plan.to_bytes()
    gpu_ops.query_select(seg, spec, 0, n, out_rows, out_ctr, kg=kg)
    torch.cuda.synchronize()
    cnt = min(int(out_ctr.item()), cap)
    return out_rows[:cnt]


def execute_grouped_values(plan: Plan, segments, metas, device: str,
                           kg=None):
    """-> (uniq_keys [g, nk] int64 cpu, per-meta dict
    {mi: (sorted_vals float64 cpu, starts, counts)}) for quantile
    finishing. Raises ValueError for unsupported key/metric families."""
    import copy
    sub = copy.deepcopy(plan)
    sub.select_rows = True
    key_cols = [[] for _ in plan.keys]
    val_cols = [[] for _ in metas]
    for seg in segments:
        if seg.n_rows == 0:
            continue
        rows_t = _matching_rows_t(sub, seg, device, kg=kg)
        if rows_t.numel() == 0:
            continue
        if rows_t.device != seg.u64.device:
            rows_t = rows_t.to(seg.u64.device)
        for ki, k in enumerate(plan.keys):
            key_cols[ki].append(_col_torch(seg, k.family, k.idx, k.bucket,
                                           plan.time_base_s, rows_t,
                                           kg=kg).cpu())
        for mi, meta in enumerate(metas):
            val_cols[mi].append(_col_torch(seg, meta["family"],
                                           meta["idx"], 0,
                                           plan.time_base_s, rows_t,
                                           kg=kg).cpu())
    if plan.keys and not key_cols[0]:
        return torch.zeros((0, len(plan.keys)), dtype=torch.int64), {}
    if plan.keys:
        keys = torch.stack([torch.cat(c) for c in key_cols], dim=1)
        uniq, inverse = torch.unique(keys, dim=0, return_inverse=True)
    else:
        n_total = sum(int(t.numel()) for t in val_cols[0]) if metas else 0
        uniq = torch.zeros((1, 0), dtype=torch.int64)
        inverse = torch.zeros(n_total, dtype=torch.int64)
    g = uniq.shape[0]
    counts = torch.bincount(inverse, minlength=g)
    starts = torch.cumsum(counts, 0) - counts
    out = {}
    _VBITS = 44  # composite sort key: (group << 44) | value
    _VMAX = (1 << _VBITS) - 1
    for mi in range(len(metas)):
        vals64 = torch.cat(val_cols[mi])
        if device != "cpu" and torch.cuda.is_available() and \
                vals64.numel() and int(vals64.max()) <= _VMAX and \
                g < (1 << (64 - _VBITS)):
            # device path: ONE rocPRIM radix sort of the composite key
            # replaces the two torch.argsort passes
            from ..ops import gpu_ops
            comp = (inverse.to("cuda") << _VBITS) | \
                vals64.to("cuda").clamp_(min=0)
            gpu_ops.sort_u64(comp)
            torch.cuda.synchronize()
            svals = (comp & _VMAX).cpu().to(torch.float64)
        else:
            vals = vals64.to(torch.float64)
            order1 = torch.argsort(vals)
            inv1 = inverse[order1]
            order2 = torch.argsort(inv1, stable=True)
            svals = vals[order1[order2]]
        out[mi] = (svals, starts, counts)
    return uniq, out
