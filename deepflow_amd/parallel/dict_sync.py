"""Cross-shard SmartEncoding dictionary synchronization.

Each GPU shard interns tag strings locally (slot-index IDs are shard-local).
After each ingest step, ranks exchange their newly discovered
(domain, id, string) entries so any rank can hydrate any shard's rows at
query time. This replaces the reference's controller-pushed dictionary
versioning (grpc_platformdata.go ReloadMaster / GetPrometheusLabelIDs) with
a collective over xGMI.

Transport: the delta strings are serialized to a byte tensor and exchanged
with all_gather (RCCL on cuda, gloo on cpu) — small after warmup, so a
direct all-gather beats any ring scheme. Steady-state cost is one ~16-byte
all_gather of the per-rank delta sizes.
"""
from __future__ import annotations

import struct
from typing import Dict, List, Tuple

import torch
import torch.distributed as dist


def _pack(entries: List[Tuple[int, int, bytes]]) -> bytes:
    parts = [struct.pack("<I", len(entries))]
    for dom, ident, s in entries:
        parts.append(struct.pack("<BIH", dom, ident, len(s)))
        parts.append(s)
    return b"".join(parts)


def _unpack(data: bytes) -> List[Tuple[int, int, bytes]]:
    (n,) = struct.unpack_from("<I", data, 0)
    pos = 4
    out = []
    for _ in range(n):
        dom, ident, ln = struct.unpack_from("<BIH", data, pos)
        pos += 7
        out.append((dom, ident, data[pos:pos + ln]))
        pos += ln
    return out


class DictSync:
    def __init__(self, dictionary, device: str | None = None,
                 max_remote_entries: int = 1 << 18):
        self.dict = dictionary
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self.device = device or ("cuda" if dist.get_backend() == "nccl" else "cpu")
        # remote hydration maps: rank -> {(dom, id): bytes}. BOUNDED:
        # with data-plane span routing each row lives on the shard that
        # interned its strings, so remote hydration is a query-merge
        # nicety, not a full mirror — FIFO-evict past the cap instead of
        # duplicating 8 shards' dictionaries in host memory (round-1
        # unbounded-growth defect).
        from collections import OrderedDict
        self.max_remote_entries = max_remote_entries
        self.remote: Dict[int, "OrderedDict[Tuple[int, int], bytes]"] = {
            r: OrderedDict() for r in range(self.world)}
        self.remote_evicted = 0
        self.bytes_exchanged = 0

    def sync_step(self) -> int:
        """Exchange pending deltas; returns total remote entries merged."""
        mine = self.dict.pending_sync
        self.dict.pending_sync = []
        payload = _pack(mine)
        sizes = torch.zeros(self.world, dtype=torch.int64, device=self.device)
        sizes[self.rank] = len(payload)
        dist.all_reduce(sizes)
        max_size = int(sizes.max().item())
        if max_size == 0:
            return 0
        buf = torch.zeros(max_size, dtype=torch.uint8, device=self.device)
        if payload:
            buf[: len(payload)] = torch.frombuffer(
                bytearray(payload), dtype=torch.uint8).to(self.device)
        gathered = [torch.zeros(max_size, dtype=torch.uint8,
                                device=self.device)
                    for _ in range(self.world)]
        dist.all_gather(gathered, buf)
        merged = 0
        for r in range(self.world):
            if r == self.rank:
                continue
            size = int(sizes[r].item())
            if size == 0:
                continue
            entries = _unpack(gathered[r][:size].cpu().numpy().tobytes())
            table = self.remote[r]
            for dom, ident, s in entries:
                table[(dom, ident)] = s
            while len(table) > self.max_remote_entries:
                table.popitem(last=False)
                self.remote_evicted += 1
            merged += len(entries)
            self.bytes_exchanged += size
        return merged

    def hydrate_remote(self, rank: int, dom: int, ident: int):
        if rank == self.rank:
            return self.dict.lookup_str(dom, ident)
        return self.remote.get(rank, {}).get((dom, ident))
