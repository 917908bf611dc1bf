"""Native RCCL communicator binding — the Horovod-core (C++) data plane.

SURVEY.md §2.5 "Horovod core" row: the reference's collective engine is C++
(fusion buffer + NCCL). This module binds ``libddlw_rccl.so``
(``parallel/hip/rccl_comm.cpp``) with ctypes and runs ddlw's bucketed
gradient all-reduce directly on an RCCL communicator owned by this layer:

- **bootstrap** (C1): rank 0 calls ``ncclGetUniqueId``; the id bytes travel
  over the already-initialised ``torch.distributed`` store
  (``broadcast_object_list``) — no MPI, matching the single-node
  spawn-rendezvous design;
- **data plane** (C2/C3): allreduce uses RCCL's fused ``ncclAvg`` (one pass,
  no separate divide kernel) enqueued on a dedicated **side HIP stream**;
  ordering against compute is by HIP events both ways, so collectives
  overlap the remaining backward exactly like Horovod's background thread;
- xGMI notes as in :mod:`ddlw_amd.parallel.api`: buckets default to 32 MiB so
  each of the 7 per-peer mesh shards stays bandwidth-bound.

Enabled with ``DDLW_NATIVE_CC=1`` (the default data plane is
``torch.distributed``'s ProcessGroupNCCL, which is also RCCL on ROCm — both
paths are RCCL over xGMI; this one removes the torch collectives layer).
"""
from __future__ import annotations

import ctypes
import os
from pathlib import Path
from typing import Optional

import torch

_LIB_PATH = Path(__file__).resolve().parent / "libddlw_rccl.so"
_lib: Optional[ctypes.CDLL] = None

_DTYPE_CODE = {
    torch.float32: 0,
    torch.bfloat16: 1,
    torch.float64: 2,
    torch.int64: 3,
    torch.uint8: 4,
    torch.float16: 5,
    torch.int32: 6,
}


def _load() -> ctypes.CDLL:
    global _lib
    if _lib is not None:
        return _lib
    if not _LIB_PATH.exists():
        raise RuntimeError(
            f"native RCCL library missing: {_LIB_PATH} — build it with "
            "`python -m ddlw_amd.parallel.build`"
        )
    lib = ctypes.CDLL(str(_LIB_PATH))
    lib.ddlw_rccl_unique_id_bytes.restype = ctypes.c_int
    lib.ddlw_rccl_get_unique_id.argtypes = [ctypes.c_char_p]
    lib.ddlw_rccl_get_unique_id.restype = ctypes.c_int
    lib.ddlw_rccl_comm_init.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_char_p]
    lib.ddlw_rccl_comm_init.restype = ctypes.c_longlong
    lib.ddlw_rccl_comm_destroy.argtypes = [ctypes.c_longlong]
    lib.ddlw_rccl_comm_destroy.restype = ctypes.c_int
    lib.ddlw_rccl_allreduce.argtypes = [
        ctypes.c_longlong, ctypes.c_void_p, ctypes.c_longlong,
        ctypes.c_int, ctypes.c_int, ctypes.c_void_p,
    ]
    lib.ddlw_rccl_allreduce.restype = ctypes.c_int
    lib.ddlw_rccl_broadcast.argtypes = [
        ctypes.c_longlong, ctypes.c_void_p, ctypes.c_longlong,
        ctypes.c_int, ctypes.c_int, ctypes.c_void_p,
    ]
    lib.ddlw_rccl_broadcast.restype = ctypes.c_int
    lib.ddlw_rccl_last_error.restype = ctypes.c_char_p
    _lib = lib
    return lib


def available() -> bool:
    return _LIB_PATH.exists()


def enabled() -> bool:
    """Native data plane requested via env (default: torch.distributed)."""
    return os.environ.get("DDLW_NATIVE_CC", "0") == "1"


def _check(rc: int, what: str) -> None:
    if rc != 0:
        err = _load().ddlw_rccl_last_error().decode()
        raise RuntimeError(f"RCCL {what} failed (rc={rc}): {err}")


class NativeComm:
    """One RCCL communicator (one process per GPU) + a side collective stream."""

    def __init__(self, nranks: int, rank: int, device: torch.device):
        lib = _load()
        self.nranks = nranks
        self.rank = rank
        self.device = device
        torch.cuda.set_device(device)
        nbytes = lib.ddlw_rccl_unique_id_bytes()
        if nranks > 1:
            import torch.distributed as dist

            if rank == 0:
                buf = ctypes.create_string_buffer(nbytes)
                _check(lib.ddlw_rccl_get_unique_id(buf), "get_unique_id")
                obj = [bytes(buf.raw)]
            else:
                obj = [None]
            # id bytes ride the existing torch.distributed rendezvous (C1)
            dist.broadcast_object_list(obj, src=0)
            id_bytes = obj[0]
        else:
            buf = ctypes.create_string_buffer(nbytes)
            _check(lib.ddlw_rccl_get_unique_id(buf), "get_unique_id")
            id_bytes = bytes(buf.raw)
        self._comm = lib.ddlw_rccl_comm_init(nranks, rank, id_bytes)
        if self._comm == 0:
            err = lib.ddlw_rccl_last_error().decode()
            raise RuntimeError(f"ncclCommInitRank failed: {err}")
        self.stream = torch.cuda.Stream(device=device)

    # ------------------------------------------------------------------ #
    def allreduce_(self, tensor: torch.Tensor, average: bool = True) -> torch.cuda.Event:
        """In-place async all-reduce on the side stream; returns the event a
        consumer stream must wait on. The producing (current) stream is
        waited on first, so the tensor is complete before RCCL reads it."""
        assert tensor.is_cuda and tensor.is_contiguous()
        lib = _load()
        ready = torch.cuda.Event()
        ready.record(torch.cuda.current_stream(self.device))
        self.stream.wait_event(ready)
        _check(
            lib.ddlw_rccl_allreduce(
                self._comm, tensor.data_ptr(), tensor.numel(),
                _DTYPE_CODE[tensor.dtype], 1 if average else 0,
                ctypes.c_void_p(self.stream.cuda_stream),
            ),
            "allreduce",
        )
        done = torch.cuda.Event()
        done.record(self.stream)
        return done

    def broadcast_(self, tensor: torch.Tensor, root: int = 0) -> None:
        """In-place broadcast on the current stream (init-time path)."""
        assert tensor.is_cuda and tensor.is_contiguous()
        lib = _load()
        cur = torch.cuda.current_stream(self.device)
        _check(
            lib.ddlw_rccl_broadcast(
                self._comm, tensor.data_ptr(), tensor.numel(),
                _DTYPE_CODE[tensor.dtype], root,
                ctypes.c_void_p(cur.cuda_stream),
            ),
            "broadcast",
        )

    def destroy(self) -> None:
        if getattr(self, "_comm", 0):
            _load().ddlw_rccl_comm_destroy(self._comm)
            self._comm = 0

    def __del__(self):  # best-effort; explicit destroy preferred
        try:
            self.destroy()
        except Exception:
            pass
