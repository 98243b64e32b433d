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


def _seg_state(seg) -> Dict:
    n = seg.n_rows
    out = {"n_rows": n, "capacity": seg.capacity,
           "pool_len": seg.pool_len,
           "attr_pool_len": getattr(seg, "attr_pool_len", 0)}
    for name in ("u64", "u32", "u8", "did", "kg", "str_lens"):
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
    for name in ("u64", "u32", "u8", "did", "kg", "str_lens"):
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


def save_l7(pipeline, path: str) -> None:
    segs = pipeline.segments
    payload = {
        "layout_version": S.LAYOUT_VERSION,
        "time_base_s": pipeline.time_base_s,
        "segment_rows": segs.segment_rows,
        "segments": [_seg_state(s) for s in segs.segments],
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
        "metrics": pipeline.metrics.state_dict()
        if hasattr(pipeline.metrics, "state_dict") else None,
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
    total = 0
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
    if payload.get("metrics") and hasattr(pipeline.metrics,
                                          "load_state_dict"):
        pipeline.metrics.load_state_dict(payload["metrics"])
    return total
