"""Checkpoint / restore of the HBM hot store.

Serializes the shard's queryable state — segments (hot + compressed
cold), the SmartEncoding dictionary (device slot table + host maps), the
knowledge-graph table and the 1s rollups — so a server restart resumes
with its history intact. The manifest carries the column-layout version;
registered migrations upgrade old checkpoints on load (the reference's
ckissu schema-migration role, server/controller/db/migrator).

Format: one torch.save archive of CPU tensors + plain dicts per shard.
"""
from __future__ import annotations

import os
from typing import Callable, Dict

import torch

from . import l7_schema as S

# version -> migration fn(payload_dict) -> payload_dict (one step up)
MIGRATIONS: Dict[int, Callable[[Dict], Dict]] = {}


def register_migration(from_version: int):
    def deco(fn):
        MIGRATIONS[from_version] = fn
        return fn
    return deco


@register_migration(4)
def _v4_to_v5(payload: Dict) -> Dict:
    """v5 added three binary trace/span id u64 columns: old rows keep the
    ids in the string pool (the fallback form), so the new columns pad
    with zeros (= 'use the pool fallback')."""
    for st in payload.get("segments", []):
        u = st["u64"]
        z = torch.zeros((3, u.shape[1]), dtype=u.dtype)
        st["u64"] = torch.cat([u, z], 0)
    for st in payload.get("cold", []):
        n = st["n_rows"]
        for _ in range(3):
            st["u64_cols"].append({"base": 0, "bits": 0, "data": None,
                                   "raw": None, "n": n})
        st["layout_version"] = 5
    payload["layout_version"] = 5
    return payload


@register_migration(3)
def _v3_to_v4(payload: Dict) -> Dict:
    """v4 dropped the per-row KG block (query-time join): discard the
    saved kg tensors; the KG table itself still restores."""
    for st in payload.get("segments", []):
        st.pop("kg", None)
    for st in payload.get("cold", []):
        st.pop("kg_cols", None)
        st["layout_version"] = 4
    payload["layout_version"] = 4
    return payload


@register_migration(2)
def _v2_to_v3(payload: Dict) -> Dict:
    """v3 appended two pooled ip6 columns to the string block: pad the
    per-column length tensor with zero rows (old rows have no ip6)."""
    def pad(sl):
        z = torch.zeros((2, sl.shape[1]), dtype=sl.dtype)
        return torch.cat([sl, z], 0)
    for st in payload.get("segments", []):
        st["str_lens"] = pad(st["str_lens"])
    for st in payload.get("cold", []):
        st["str_lens"] = pad(st["str_lens"])
        st["layout_version"] = 3
    payload["layout_version"] = 3
    return payload


def _seg_state(seg) -> Dict:
    n = seg.n_rows
    out = {"n_rows": n, "capacity": seg.capacity,
           "pool_len": seg.pool_len,
           "attr_pool_len": getattr(seg, "attr_pool_len", 0)}
    for name in ("u64", "u32", "u8", "did", "str_lens"):
        t = getattr(seg, name, None)
        if t is not None:
            out[name] = t[..., :n].cpu().clone()
    out["str_rowref"] = seg.str_rowref[:n].cpu().clone()
    out["pool"] = seg.pool[: seg.pool_len].cpu().clone()
    if hasattr(seg, "attr_start"):
        out["attr_start"] = seg.attr_start[:n].cpu().clone()
        out["attr_cnt"] = seg.attr_cnt[:n].cpu().clone()
        out["attr_pool"] = seg.attr_pool[: seg.attr_pool_len].cpu().clone()
    return out


def _seg_restore(state: Dict, seg) -> None:
    n = state["n_rows"]
    for name in ("u64", "u32", "u8", "did", "str_lens"):
        if name in state:
            getattr(seg, name)[..., :n] = state[name].to(seg.device)
    seg.str_rowref[:n] = state["str_rowref"].to(seg.device)
    seg.ensure_pool(state["pool_len"])
    seg.pool[: state["pool_len"]] = state["pool"].to(seg.device)
    seg.pool_len = state["pool_len"]
    if "attr_start" in state:
        seg.attr_start[:n] = state["attr_start"].to(seg.device)
        seg.attr_cnt[:n] = state["attr_cnt"].to(seg.device)
        seg.ensure_attr_pool(state["attr_pool_len"])
        seg.attr_pool[: state["attr_pool_len"]] = \
            state["attr_pool"].to(seg.device)
        seg.attr_pool_len = state["attr_pool_len"]
    seg.n_rows = n


def _cold_state(c) -> Dict:
    """Serialize a CompressedL7Segment (packed payloads stay packed)."""
    def cols(lst):
        return [{"base": pc.base, "bits": pc.bits, "n": pc.n,
                 "data": pc.data.cpu().clone() if pc.data is not None
                 else None,
                 "raw": pc.raw.cpu().clone() if pc.raw is not None
                 else None} for pc in lst]
    return {
        "time_min": getattr(c, "time_min", 0),
        "time_max": getattr(c, "time_max", None),"n_rows": c.n_rows, "capacity": c.capacity,
            "layout_version": c.layout_version,
            "u64_cols": cols(c.u64_cols), "u32_cols": cols(c.u32_cols),
            "did_cols": cols(c.did_cols),
            "rowref_col": cols(c.rowref_col),
            "u8": c.u8.cpu().clone(), "str_lens": c.str_lens.cpu().clone(),
            "attr_start": c.attr_start.cpu().clone()
            if c.attr_start is not None else None,
            "attr_cnt": c.attr_cnt.cpu().clone()
            if c.attr_cnt is not None else None,
            "attr_pool": c.attr_pool.cpu().clone()
            if c.attr_pool is not None else None,
            "attr_pool_len": c.attr_pool_len,
            "pool": c.pool.cpu().clone(), "pool_len": c.pool_len}


def _cold_restore(state: Dict, device: str):
    from .coldstore import CompressedL7Segment, PackedColumn
    c = CompressedL7Segment.__new__(CompressedL7Segment)

    def cols(lst):
        return [PackedColumn(
            d["base"], d["bits"],
            d["data"].to(device) if d["data"] is not None else None,
            d["n"],
            raw=d["raw"].to(device) if d["raw"] is not None else None)
            for d in lst]
    c.n_rows = state["n_rows"]
    c.capacity = state["capacity"]
    c.device = device
    c.layout_version = state["layout_version"]
    c.u64_cols = cols(state["u64_cols"])
    c.u32_cols = cols(state["u32_cols"])
    c.did_cols = cols(state["did_cols"])
    c.rowref_col = cols(state["rowref_col"])
    for name in ("u8", "str_lens", "attr_start", "attr_cnt", "attr_pool",
                 "pool"):
        t = state[name]
        setattr(c, name, t.to(device) if t is not None else None)
    c.attr_pool_len = state["attr_pool_len"]
    c.pool_len = state["pool_len"]
    c.time_min = state.get("time_min", 0)
    c.time_max = state.get("time_max", None)
    return c


def save_l7(pipeline, path: str) -> None:
    segs = pipeline.segments
    payload = {
        "layout_version": S.LAYOUT_VERSION,
        "time_base_s": pipeline.time_base_s,
        "segment_rows": segs.segment_rows,
        "segments": [_seg_state(s) for s in segs.segments],
        "cold": [_cold_state(c) for c in getattr(segs, "cold", [])],
        "dict": {
            "capacity": pipeline.dict.capacity,
            "tkeys": pipeline.dict.tkeys.cpu().clone(),
            "id_to_str": dict(pipeline.dict.id_to_str),
        },
        "kg": {
            "tkeys": pipeline.kg.tkeys.cpu().clone(),
            "tvals": pipeline.kg.tvals.cpu().clone(),
            "host": dict(pipeline.kg.host),
            "version": pipeline.kg.version,
        },
        "rollups": pipeline.rollups.state_dict()
        if hasattr(pipeline, "rollups") else None,
    }
    tmp = path + ".tmp"
    torch.save(payload, tmp)
    os.replace(tmp, path)


def load_l7(pipeline, path: str) -> int:
    """Restore into a fresh pipeline (same segment_rows/device). Applies
    layout migrations as needed. Returns rows restored."""
    payload = torch.load(path, map_location="cpu", weights_only=False)
    v = payload.get("layout_version", 0)
    while v < S.LAYOUT_VERSION:
        mig = MIGRATIONS.get(v)
        if mig is None:
            raise RuntimeError(
                f"no migration from layout v{v} to v{S.LAYOUT_VERSION}")
        payload = mig(payload)
        v = payload["layout_version"]
    segs = pipeline.segments
    # restore the time base: rows carry absolute ns timestamps, but query
    # buckets are relative to time_base_s — a mismatched base silently
    # shifts every timestamp in results
    if "time_base_s" in payload:
        pipeline.time_base_s = payload["time_base_s"]
    saved_rows = payload.get("segment_rows")
    if saved_rows is not None and saved_rows > segs.segment_rows:
        raise RuntimeError(
            f"checkpoint segment_rows={saved_rows} exceeds pipeline "
            f"segment_rows={segs.segment_rows}; rebuild the pipeline with "
            f"segment_rows>={saved_rows} to restore")
    d_cap = payload["dict"]["tkeys"].numel()
    if d_cap != pipeline.dict.tkeys.numel():
        raise RuntimeError(
            f"checkpoint dictionary capacity {d_cap} != pipeline "
            f"{pipeline.dict.tkeys.numel()}; slot ids are positional — "
            f"restore requires the same dict capacity")
    total = 0
    if payload.get("cold"):
        segs.cold = [_cold_restore(st, pipeline.device)
                     for st in payload["cold"]]
        total += sum(c.n_rows for c in segs.cold)
    for st in payload["segments"]:
        seg = segs.tail(min_free=segs.segment_rows)  # fresh segment
        _seg_restore(st, seg)
        total += st["n_rows"]
    d = payload["dict"]
    pipeline.dict.tkeys.copy_(d["tkeys"].to(pipeline.dict.tkeys.device))
    pipeline.dict.id_to_str.update(d["id_to_str"])
    for (dom, slot), sb in d["id_to_str"].items():
        pipeline.dict.str_to_id[(dom, sb)] = slot
    kg = payload["kg"]
    pipeline.kg.tkeys.copy_(kg["tkeys"].to(pipeline.kg.tkeys.device))
    pipeline.kg.tvals.copy_(kg["tvals"].to(pipeline.kg.tvals.device))
    pipeline.kg.host.update(kg["host"])
    pipeline.kg.version = kg["version"]
    if payload.get("rollups") and hasattr(pipeline, "rollups"):
        pipeline.rollups.load_state_dict(payload["rollups"])
    elif payload.get("metrics"):  # pre-table-family checkpoints
        pipeline.metrics.load_state_dict(payload["metrics"])
    return total


# ---------------------------------------------------------------- L4
def save_l4(pipeline, path: str) -> None:
    """Persist the L4 (flow) hot store. L4 segments carry no dictionary
    ids or attr pools; the shared TagDictionary/KG are saved by save_l7
    (the reference's durable-state analog is ClickHouse flow_log.l4)."""
    from . import l4_schema as L4
    segs = pipeline.segments

    def seg_state(seg):
        return {
            "n_rows": seg.n_rows,
            "capacity": seg.capacity,
            "u64": seg.u64.cpu().clone(),
            "u32": seg.u32.cpu().clone(),
            "u8": seg.u8.cpu().clone(),
            "str_rowref": seg.str_rowref.cpu().clone(),
            "str_lens": seg.str_lens.cpu().clone(),
            "pool": seg.pool[:seg.pool_len].cpu().clone(),
            "pool_len": seg.pool_len,
        }
    payload = {
        "layout_version": getattr(L4, "LAYOUT_VERSION", 1),
        "time_base_s": pipeline.time_base_s,
        "segment_rows": segs.segment_rows,
        "segments": [seg_state(s) for s in segs.segments],
        "rollups": pipeline.rollups.state_dict()
        if hasattr(pipeline, "rollups") else None,
    }
    tmp = path + ".tmp"
    torch.save(payload, tmp)
    os.replace(tmp, path)


def load_l4(pipeline, path: str) -> int:
    from . import l4_schema as L4
    state = torch.load(path, weights_only=False)
    if state["layout_version"] != getattr(L4, "LAYOUT_VERSION", 1):
        raise ValueError(
            f"l4 checkpoint layout {state['layout_version']} != "
            f"{getattr(L4, 'LAYOUT_VERSION', 1)} (no migration defined)")
    if state["time_base_s"] != pipeline.time_base_s:
        raise ValueError("time_base_s mismatch on l4 restore")
    segs = pipeline.segments
    dev = segs.device
    segs.segments = []
    total = 0
    for st in state["segments"]:
        seg = segs.cls(st["capacity"], device=dev)
        for name in ("u64", "u32", "u8", "str_rowref", "str_lens"):
            getattr(seg, name).copy_(st[name].to(dev))
        seg.ensure_pool(st["pool_len"])
        if st["pool_len"]:
            seg.pool[:st["pool_len"]] = st["pool"].to(dev)
        seg.pool_len = st["pool_len"]
        seg.n_rows = st["n_rows"]
        segs.segments.append(seg)
        total += seg.n_rows
    if hasattr(pipeline, "rollups") and state.get("rollups") is not None:
        pipeline.rollups.load_state_dict(state["rollups"])
    return total
