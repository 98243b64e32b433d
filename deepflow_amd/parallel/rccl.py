"""Direct librccl C-API bindings for the data-plane collectives.

VERDICT r1 #8: shrink the torch surface — torch.distributed stays for
bootstrap/control (rendezvous of the ncclUniqueId over gloo), but the
hot-path byte exchanges (span-routing all-to-all, dictionary-delta
all-gather) can run straight on librccl: ncclGroup'd ncclSend/ncclRecv
pairs on our stream, no ProcessGroup machinery in the timed path.
Enable with DF_RCCL_DIRECT=1 (SpanRouter/DictSync pick it up); the
torch.distributed path remains the default transport.

librccl.so ships inside the torch wheel (torch.distributed's "nccl"
backend IS this library on ROCm) — we bind the same .so directly.
"""
from __future__ import annotations

import ctypes as ct
import os
from typing import List, Optional

import torch

NCCL_UNIQUE_ID_BYTES = 128
ncclSuccess = 0
ncclInt8 = 0  # ncclDataType_t


class NcclUniqueId(ct.Structure):
    # passed BY VALUE to ncclCommInitRank (a bare ctypes char array
    # would decay to a pointer and the init fails with invalid argument)
    _fields_ = [("internal", ct.c_char * NCCL_UNIQUE_ID_BYTES)]


def _find_librccl() -> str:
    cand = os.path.join(os.path.dirname(torch.__file__), "lib",
                        "librccl.so")
    if os.path.exists(cand):
        return cand
    return "librccl.so"


_lib: Optional[ct.CDLL] = None


def lib() -> ct.CDLL:
    global _lib
    if _lib is None:
        h = ct.CDLL(_find_librccl(), mode=ct.RTLD_GLOBAL)
        h.ncclGetUniqueId.restype = ct.c_int
        h.ncclGetUniqueId.argtypes = [ct.c_void_p]
        h.ncclCommInitRank.restype = ct.c_int
        h.ncclCommInitRank.argtypes = [ct.c_void_p, ct.c_int,
                                       NcclUniqueId, ct.c_int]
        h.ncclCommDestroy.restype = ct.c_int
        h.ncclCommDestroy.argtypes = [ct.c_void_p]
        h.ncclGroupStart.restype = ct.c_int
        h.ncclGroupEnd.restype = ct.c_int
        h.ncclSend.restype = ct.c_int
        h.ncclSend.argtypes = [ct.c_void_p, ct.c_size_t, ct.c_int,
                               ct.c_int, ct.c_void_p, ct.c_void_p]
        h.ncclRecv.restype = ct.c_int
        h.ncclRecv.argtypes = [ct.c_void_p, ct.c_size_t, ct.c_int,
                               ct.c_int, ct.c_void_p, ct.c_void_p]
        h.ncclAllGather.restype = ct.c_int
        h.ncclAllGather.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_size_t,
                                    ct.c_int, ct.c_void_p, ct.c_void_p]
        h.ncclGetErrorString.restype = ct.c_char_p
        h.ncclGetErrorString.argtypes = [ct.c_int]
        _lib = h
    return _lib


def _check(rc: int, what: str) -> None:
    if rc != ncclSuccess:
        raise RuntimeError(
            f"rccl error in {what}: "
            f"{lib().ncclGetErrorString(rc).decode()}")


def enabled() -> bool:
    return bool(os.environ.get("DF_RCCL_DIRECT"))


class RcclComm:
    """One communicator per process (rank = torch.distributed rank),
    bootstrapped by broadcasting the unique id over the existing
    torch.distributed group (gloo or nccl — control plane only)."""

    def __init__(self):
        import torch.distributed as dist
        h = lib()
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        uid = NcclUniqueId()
        if self.rank == 0:
            _check(h.ncclGetUniqueId(ct.byref(uid)), "ncclGetUniqueId")
        # c_char fields NUL-truncate on attribute access; take raw bytes
        blob = [ct.string_at(ct.byref(uid), NCCL_UNIQUE_ID_BYTES)]
        dist.broadcast_object_list(blob, src=0)
        uid = NcclUniqueId.from_buffer_copy(blob[0])
        self.comm = ct.c_void_p()
        _check(h.ncclCommInitRank(ct.byref(self.comm), self.world, uid,
                                  self.rank), "ncclCommInitRank")

    def _stream(self) -> int:
        return torch.cuda.current_stream().cuda_stream

    def all_to_all_u8(self, send: torch.Tensor, send_splits: List[int],
                      recv: torch.Tensor, recv_splits: List[int]) -> None:
        """Direct-send all-to-all of byte buffers (xGMI is point-to-point
        — pairwise sends are per-link optimal, no ring)."""
        h = lib()
        _check(h.ncclGroupStart(), "ncclGroupStart")
        soff = roff = 0
        for peer in range(self.world):
            if send_splits[peer]:
                _check(h.ncclSend(send.data_ptr() + soff,
                                  send_splits[peer], ncclInt8, peer,
                                  self.comm, self._stream()), "ncclSend")
            soff += send_splits[peer]
            if recv_splits[peer]:
                _check(h.ncclRecv(recv.data_ptr() + roff,
                                  recv_splits[peer], ncclInt8, peer,
                                  self.comm, self._stream()), "ncclRecv")
            roff += recv_splits[peer]
        _check(h.ncclGroupEnd(), "ncclGroupEnd")

    def all_gather_u8(self, send: torch.Tensor,
                      recv: torch.Tensor) -> None:
        """recv[world * len(send)] <- every rank's send buffer."""
        _check(lib().ncclAllGather(send.data_ptr(), recv.data_ptr(),
                                   send.numel(), ncclInt8, self.comm,
                                   self._stream()), "ncclAllGather")

    def close(self) -> None:
        if self.comm:
            lib().ncclCommDestroy(self.comm)
            self.comm = None


_comm: Optional[RcclComm] = None


def comm() -> RcclComm:
    global _comm
    if _comm is None:
        _comm = RcclComm()
    return _comm
