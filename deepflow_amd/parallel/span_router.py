"""Data-plane shard routing: every span lands on its owner GPU.

The round-1 gap: ranks ingested disjoint generator streams —
nothing routed a *received* record to the shard that owns its agent. This
module is that path: peek each record's vtap_id natively
(df_route_spans — the reference hashes frames to decode queues by agent
ip, server/libs/receiver/receiver.go:519-566), gather per-destination
packed buffers (GPU: k_gather_records), and exchange them with ONE
all_to_all_single over RCCL/xGMI per batch (direct send — xGMI is 7
point-to-point links per GPU, so all-to-all is per-link optimal; no ring).
Received buffers feed the local pipeline unchanged.

gloo (CPU tests / no GPU) lacks all_to_all: the exchange falls back to
all_gather_object with receiver-side selection — same routing semantics,
test-only transport.
"""
from __future__ import annotations

import ctypes as ct
from typing import List, Optional, Tuple

import numpy as np
import torch
import torch.distributed as dist

from ..ops import native


def shard_of(payload: np.ndarray, offs: np.ndarray, lens: np.ndarray,
             world: int, org_id: int = 1) -> np.ndarray:
    """Owner shard per record from base.vtap_id (native peek)."""
    lib = native.cpu()
    out = np.zeros(len(offs), dtype=np.uint8)
    lib.df_route_spans(payload.ctypes.data_as(ct.c_void_p), len(payload),
                       offs.ctypes.data_as(ct.c_void_p),
                       lens.ctypes.data_as(ct.c_void_p), len(offs),
                       world, org_id,
                       out.ctypes.data_as(ct.c_void_p))
    return out


class SpanRouter:
    def __init__(self, device: str = "cpu", org_id: int = 1):
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self.device = device
        self.org_id = org_id
        self.bytes_sent = 0
        self.records_routed = 0

    # ------------------------------------------------------------ common
    def _split(self, payload: np.ndarray, offs: np.ndarray,
               lens: np.ndarray):
        """-> (order, per-dest record counts, per-dest byte counts,
        packed dst offsets for the gathered order)."""
        shard = shard_of(payload, offs, lens, self.world, self.org_id)
        order = np.argsort(shard, kind="stable").astype(np.uint32)
        sorted_shard = shard[order]
        counts = np.bincount(sorted_shard,
                             minlength=self.world).astype(np.int64)
        sel_lens = lens[order].astype(np.int64)
        dst_off = np.zeros(len(order), dtype=np.uint64)
        if len(order) > 1:
            dst_off[1:] = np.cumsum(sel_lens[:-1]).view(np.int64)
        byte_counts = np.bincount(
            sorted_shard, weights=lens.astype(np.float64),
            minlength=self.world).astype(np.int64)  # exact below 2^53
        return order, counts, byte_counts, dst_off, sel_lens

    # -------------------------------------------------------------- gpu
    def route_gpu(self, payload_t: torch.Tensor, offs_t: torch.Tensor,
                  lens_t: torch.Tensor, payload_host: np.ndarray,
                  offs_host: np.ndarray, lens_host: np.ndarray):
        """RCCL path: pack per-destination buffers on device, exchange
        with all_to_all_single, return (payload_t, offs_t, lens_t,
        payload_host_mine) of the records this shard owns."""
        from ..ops import gpu_ops
        dev = payload_t.device
        order, counts, byte_counts, dst_off, sel_lens = self._split(
            payload_host, offs_host, lens_host)
        m = len(order)
        sel_t = torch.from_numpy(order.view(np.int32)).to(dev)
        dst_off_t = torch.from_numpy(dst_off.view(np.int64)).to(dev)
        packed = torch.empty(int(sel_lens.sum()), dtype=torch.uint8,
                             device=dev)
        gpu_ops.gather_records(payload_t, offs_t, lens_t, sel_t, dst_off_t,
                               packed)
        from . import rccl
        if rccl.enabled():
            # direct librccl: all-gather the per-dest (count, bytes)
            # matrix, then grouped pairwise send/recv for payload + lens
            cm = rccl.comm()
            meta = np.stack([counts, byte_counts], axis=1).ravel().copy()
            meta_t = torch.from_numpy(meta.view(np.uint8)).to(dev)
            meta_all = torch.empty(self.world * meta_t.numel(),
                                   dtype=torch.uint8, device=dev)
            cm.all_gather_u8(meta_t, meta_all)
            am = meta_all.cpu().numpy().view(np.int64).reshape(
                self.world, self.world, 2)
            recv_counts = [int(am[src, self.rank, 0])
                           for src in range(self.world)]
            recv_bytes = [int(am[src, self.rank, 1])
                          for src in range(self.world)]
            out_payload = torch.empty(int(sum(recv_bytes)),
                                      dtype=torch.uint8, device=dev)
            cm.all_to_all_u8(packed, byte_counts.tolist(), out_payload,
                             recv_bytes)
            sel_lens_t = torch.from_numpy(
                sel_lens.astype(np.int32)).to(dev)
            out_lens = torch.empty(int(sum(recv_counts)),
                                   dtype=torch.int32, device=dev)
            cm.all_to_all_u8(sel_lens_t.view(torch.uint8),
                             [c * 4 for c in counts.tolist()],
                             out_lens.view(torch.uint8),
                             [c * 4 for c in recv_counts])
            torch.cuda.synchronize()
        else:
            # exchange split sizes: per-destination [count, bytes] pairs
            send_sizes = torch.from_numpy(
                np.stack([counts, byte_counts],
                         axis=1).ravel().copy()).to(dev)
            recv_sizes = torch.empty_like(send_sizes)
            dist.all_to_all_single(recv_sizes, send_sizes)
            rs = recv_sizes.view(self.world, 2).cpu().numpy()
            recv_counts = rs[:, 0].tolist()
            recv_bytes = rs[:, 1].tolist()
            # payload bytes
            out_payload = torch.empty(int(sum(recv_bytes)),
                                      dtype=torch.uint8, device=dev)
            dist.all_to_all_single(out_payload, packed,
                                   output_split_sizes=recv_bytes,
                                   input_split_sizes=byte_counts.tolist())
            # record lens
            sel_lens_t = torch.from_numpy(
                sel_lens.astype(np.int32)).to(dev)
            out_lens = torch.empty(int(sum(recv_counts)),
                                   dtype=torch.int32, device=dev)
            dist.all_to_all_single(out_lens, sel_lens_t,
                                   output_split_sizes=recv_counts,
                                   input_split_sizes=counts.tolist())
        # offsets: blocks arrive in source order, payload likewise — a
        # global exclusive cumsum of lens reconstructs offsets
        cum = torch.cumsum(out_lens.to(torch.int64), 0)
        out_offs = (cum - out_lens.to(torch.int64)).to(torch.int32)
        self.bytes_sent += int(sel_lens.sum())
        self.records_routed += m
        return out_payload, out_offs, out_lens

    # -------------------------------------------------------------- cpu
    def route_cpu(self, payload: np.ndarray, offs: np.ndarray,
                  lens: np.ndarray) -> Tuple[np.ndarray, np.ndarray,
                                             np.ndarray]:
        """gloo fallback: object all-gather of per-destination blobs."""
        shard = shard_of(payload, offs, lens, self.world, self.org_id)
        blobs: List[bytes] = []
        for d in range(self.world):
            idx = np.nonzero(shard == d)[0]
            parts = [payload[offs[i]:offs[i] + lens[i]].tobytes()
                     for i in idx]
            blobs.append((np.array([len(p) for p in parts],
                                   dtype=np.uint32).tobytes(),
                          b"".join(parts)))
        gathered: List = [None] * self.world
        dist.all_gather_object(gathered, blobs)
        lens_out: List[int] = []
        chunks: List[bytes] = []
        for src in range(self.world):
            lens_b, data = gathered[src][self.rank]
            lens_arr = np.frombuffer(lens_b, dtype=np.uint32)
            lens_out.extend(int(x) for x in lens_arr)
            chunks.append(data)
        blob = b"".join(chunks)
        out_payload = np.frombuffer(blob, dtype=np.uint8)
        out_lens = np.array(lens_out, dtype=np.uint32)
        out_offs = np.zeros(len(out_lens), dtype=np.uint32)
        if len(out_lens) > 1:
            np.cumsum(out_lens[:-1], out=out_offs[1:])
        self.records_routed += len(offs)
        return out_payload, out_offs, out_lens
