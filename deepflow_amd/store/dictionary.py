"""SmartEncoding tag dictionary (string interning).

GPU side: one open-addressing table (u64 hash keys; DICT ID == slot index)
shared by all dict domains, written by the intern kernel (K3). New entries
are emitted to a side buffer as (domain<<56|slot, packed batch str-ref); the
host harvests them after each batch into the authoritative id<->string maps
used for query-time hydration and string-literal -> id filter compilation.

This realizes the reference's SmartEncoding write path: integer IDs stored in
columns (grpc_platformdata.go / flow_tag_writer.go roles), with query-time
re-hydration replacing ClickHouse dictGet (tag/translation.go:101).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from . import l7_schema as S

EMIT_CAP = 1 << 20


def str_hash_py(data: bytes, seed: int) -> int:
    """Python twin of dfgpu.hip str_hash (used by the CPU reference ops)."""
    M = (1 << 64) - 1
    h = (seed ^ 0x27D4EB2F165667C5 ^ (len(data) * 0x9E3779B97F4A7C15)) & M
    i = 0
    n = len(data)
    while i + 8 <= n:
        k = int.from_bytes(data[i:i + 8], "little")
        h ^= (k * 0xC2B2AE3D27D4EB4F) & M
        h = ((h << 31) | (h >> 33)) & M
        h = (h * 0x9E3779B185EBCA87) & M
        i += 8
    tail = 0
    for j in range(n - i):
        tail |= data[i + j] << (8 * j)
    h ^= (tail * 0x165667B19E3779F9) & M
    h ^= h >> 33
    h = (h * 0xFF51AFD7ED558CCD) & M
    h ^= h >> 29
    h = (h * 0xC4CEB9FE1A85EC53) & M
    h ^= h >> 32
    return h if h else 1


def domain_seed(domain: int) -> int:
    return (0x9E3779B97F4A7C15 * (domain + 1)) & ((1 << 64) - 1)


class TagDictionary:
    """Per-shard tag dictionary: GPU table + host id<->string maps."""

    def __init__(self, capacity_pow2: int = 1 << 22, device: str = "cpu"):
        assert capacity_pow2 & (capacity_pow2 - 1) == 0
        self.capacity = capacity_pow2
        self.device = device
        dev = torch.device(device)
        self.tkeys = torch.zeros(capacity_pow2, dtype=torch.int64, device=dev)
        self.emit = torch.zeros((EMIT_CAP, 2), dtype=torch.int64, device=dev)
        self.emit_ctr = torch.zeros(1, dtype=torch.int32, device=dev)
        # host maps: (domain, id) -> string ; (domain, string) -> id
        self.id_to_str: Dict[Tuple[int, int], bytes] = {}
        self.str_to_id: Dict[Tuple[int, bytes], int] = {}
        # overflow count (table-full events observed at harvest)
        self.dropped = 0
        # entries discovered since the last cross-shard sync
        self.pending_sync: List[Tuple[int, int, bytes]] = []

    def n_entries(self) -> int:
        return len(self.id_to_str)

    def harvest(self, payload) -> int:
        """Drain the emit buffer after a batch. `payload` is the batch's
        raw bytes — host numpy, or a DEVICE uint8 tensor (routed batches
        have no host copy: only the emitted slices transfer, a few bytes
        per NEW dictionary entry instead of a full-batch D2H)."""
        cnt = int(self.emit_ctr[0].item())
        if cnt == 0:
            return 0
        take = min(cnt, EMIT_CAP)
        rows = self.emit[:take].cpu().numpy()
        import torch
        is_dev = isinstance(payload, torch.Tensor) and \
            payload.device.type != "cpu"
        for tag, ref in rows:
            tag = int(tag) & 0xFFFFFFFFFFFFFFFF
            ref = int(ref) & 0xFFFFFFFFFFFFFFFF
            dom = tag >> 56
            slot = tag & 0xFFFFFFFF
            off, ln = ref >> 16, ref & 0xFFFF
            if is_dev:
                sbytes = bytes(payload[off:off + ln].cpu().numpy())
            else:
                sbytes = payload[off:off + ln].tobytes()
            self.id_to_str[(dom, slot)] = sbytes
            self.str_to_id[(dom, sbytes)] = slot
            self.pending_sync.append((dom, slot, sbytes))
        if cnt > EMIT_CAP:
            self.dropped += cnt - EMIT_CAP
        self.emit_ctr.zero_()
        return take

    def lookup_id(self, domain: int, value: bytes) -> Optional[int]:
        return self.str_to_id.get((domain, value))

    def lookup_str(self, domain: int, ident: int) -> Optional[bytes]:
        if ident == S.DICT_ID_INVALID or ident < 0:
            return None
        return self.id_to_str.get((domain, ident))

    def hydrate(self, domain: int, ids) -> List[Optional[str]]:
        out = []
        for i in ids:
            i = int(i)
            if i < 0 or i == S.DICT_ID_INVALID:
                out.append(None)
            else:
                b = self.id_to_str.get((domain, i & 0xFFFFFFFF))
                out.append(b.decode("utf-8", "replace") if b is not None else None)
        return out

    # ---- cross-shard sync (parallel/dict_sync.py drives this) ----
    def export_entries(self, since: int = 0) -> List[Tuple[int, int, bytes]]:
        """All (domain, id, string) entries; `since` reserved for deltas."""
        return [(d, i, s) for (d, i), s in self.id_to_str.items()]

    def import_entries(self, entries) -> None:
        """Merge remote (domain, id, string) entries for query-side
        hydration of remote shards (ids are shard-local; callers keep these
        in per-shard maps)."""
        for d, i, s in entries:
            self.id_to_str.setdefault((d, i), s)
            self.str_to_id.setdefault((d, s), i)
