"""Loader for the in-tree HIP kernel library (``libddlw_kernels.so``).

The library is built by ``ddlw_amd/ops/build.py`` (raw ``hipcc
--offload-arch=gfx950 -shared -fPIC``) into ``ddlw_amd/ops/`` so it travels
with the repo snapshot to GPU boxes. Kernels are exposed as ``extern "C"``
functions taking raw device pointers + the current HIP stream; Python binds
them with ctypes — torch tensors supply ``data_ptr()`` and
``torch.cuda.current_stream().cuda_stream`` supplies the stream, so kernels
land on the same stream PyTorch uses (and are capturable in hipGraphs).
"""
from __future__ import annotations

import ctypes
import os
from pathlib import Path
from typing import Optional

import torch

LIB_NAME = "libddlw_kernels.so"
LIB_DIR = Path(__file__).resolve().parent


class KernelLibError(RuntimeError):
    pass


_lib: Optional[ctypes.CDLL] = None
_load_error: Optional[str] = None


def _try_load() -> Optional[ctypes.CDLL]:
    global _lib, _load_error
    if _lib is not None:
        return _lib
    path = LIB_DIR / LIB_NAME
    if not path.exists():
        _load_error = f"{path} not built (run ddlw_amd/ops/build.py)"
        return None
    try:
        _lib = ctypes.CDLL(str(path), mode=ctypes.RTLD_GLOBAL)
    except OSError as e:
        _load_error = f"failed to load {path}: {e}"
        return None
    _lib.ddlw_last_error.restype = ctypes.c_char_p
    return _lib


def has_lib() -> bool:
    return _try_load() is not None


def lib() -> Optional[ctypes.CDLL]:
    return _try_load()


def require_lib() -> ctypes.CDLL:
    """On a GPU box the HIP library is mandatory — fail loudly, never fall
    back to eager silently (round-end check: 'native code not loaded')."""
    l = _try_load()
    if l is None:
        raise KernelLibError(
            f"ddlw HIP kernel library unavailable: {_load_error}. "
            "Build it with `python -m ddlw_amd.ops.build`."
        )
    return l


def check(status: int, name: str) -> None:
    if status != 0:
        err = _lib.ddlw_last_error().decode() if _lib is not None else "?"
        raise KernelLibError(f"{name} failed (status {status}): {err}")


def current_stream_ptr() -> int:
    return torch.cuda.current_stream().cuda_stream


def on_gpu(*tensors: torch.Tensor) -> bool:
    return all(t.is_cuda for t in tensors)
