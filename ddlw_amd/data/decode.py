"""Reusable parallel image decoder — forked worker processes + shared-memory
slots, for one-shot batch maps (the inference-side twin of the streaming
loader's decode pool).

The reference decodes inference rows serially inside the pyfunc's ``predict``
(PIL, ``Part 2 .../03_pyfunc_distributed_inference.py:214-234``), which is
single-core-bound. :class:`ParallelDecoder` keeps the same
bytes-in / stacked-tensor-out contract but decodes across N forked processes
(the GIL makes thread pools top out near one core's throughput).
"""
from __future__ import annotations

import os
import queue
from typing import Callable, Optional, Sequence

import numpy as np
import torch

from .loader import _ProcDecodePool
from .preprocess import _to_bytes


def decode_resize_u8(content, img_height: int = 224, img_width: int = 224) -> np.ndarray:
    """bytes -> uint8 HWC RGB, resized only if needed. ``draft`` lets libjpeg
    decode at a reduced DCT scale when downscaling >=2x (big sources)."""
    import io

    from PIL import Image

    img = Image.open(io.BytesIO(_to_bytes(content)))
    img.draft("RGB", (img_width, img_height))
    img = img.convert("RGB")
    if img.size != (img_width, img_height):
        img = img.resize((img_width, img_height), Image.BILINEAR)
    return np.asarray(img)


class ParallelDecoder:
    """Process-parallel ``map(transform, contents) -> stacked torch.Tensor``.

    Forks ``workers`` decode processes once (construction), then serves
    arbitrary-size batches: the input list is chunked, decoded into
    shared-memory slots, and copied into one output tensor in input order.
    Falls back to in-process serial decode when forking is unavailable
    (daemonic caller) or ``workers <= 1``.
    """

    def __init__(self, transform: Callable, workers: Optional[int] = None,
                 chunk_size: int = 64, pin: bool = False):
        import multiprocessing as mp

        self.transform = transform
        self.chunk_size = int(chunk_size)
        if workers is None:
            workers = max(1, (os.cpu_count() or 2) - 1)
        if mp.current_process().daemon:
            workers = 0  # cannot fork children
        self.workers = workers
        self.pin = pin
        self._pool: Optional[_ProcDecodePool] = None

    def _ensure_pool(self, sample: np.ndarray) -> _ProcDecodePool:
        if self._pool is None:
            self._pool = _ProcDecodePool(
                self.transform, sample.shape, sample.dtype,
                self.chunk_size, self.workers, pin=self.pin)
        return self._pool

    def map(self, contents: Sequence) -> torch.Tensor:
        n = len(contents)
        if n == 0:
            raise ValueError("empty batch")
        first = self.transform(contents[0])
        if isinstance(first, torch.Tensor):
            first = first.numpy()
        first = np.asarray(first)
        if self.workers <= 1:
            out = np.empty((n,) + first.shape, dtype=first.dtype)
            out[0] = first
            for i in range(1, n):
                r = self.transform(contents[i])
                out[i] = r.numpy() if isinstance(r, torch.Tensor) else r
            return torch.from_numpy(out)

        pool = self._ensure_pool(first)
        out = np.empty((n,) + first.shape, dtype=first.dtype)
        chunks = [(s, min(s + self.chunk_size, n)) for s in range(0, n, self.chunk_size)]
        free = list(range(pool.slots))
        pending = {}  # seq -> (start, end)
        next_submit = 0
        received = 0
        done = 0
        try:
            while done < len(chunks):
                while free and next_submit < len(chunks):
                    s, e = chunks[next_submit]
                    slot = free.pop()
                    pool.task_q.put((next_submit, slot, list(contents[s:e])))
                    pending[next_submit] = (s, e)
                    next_submit += 1
                try:
                    status, seq, slot, payload = pool.res_q.get(timeout=10.0)
                except queue.Empty:
                    dead = pool.any_dead()
                    if dead:
                        raise RuntimeError(f"decode worker(s) died (pids {dead})")
                    continue
                received += 1
                if status == "err":
                    raise RuntimeError(f"decode failed in worker: {payload}")
                s, e = pending.pop(seq)
                out[s:e] = pool.view[slot, : e - s]
                free.append(slot)
                done += 1
        except BaseException:
            self._drain(pool, next_submit - received)
            raise
        return torch.from_numpy(out)

    def _drain(self, pool: _ProcDecodePool, outstanding: int) -> None:
        """Consume in-flight results after an abnormal exit so stale seq
        ids cannot be attributed to the NEXT map/imap call on this pool."""
        for _ in range(max(0, outstanding)):
            try:
                pool.res_q.get(timeout=10.0)
            except queue.Empty:
                # workers wedged or dead — the pool is unusable; rebuild it
                pool.close()
                self._pool = None
                return

    def imap(self, contents: Sequence):
        """Pipelined chunk iterator: yields decoded ``chunk_size`` batches
        IN ORDER while later chunks decode concurrently in the pool — so a
        GPU consumer overlaps forward with decode. The yielded tensor is a
        zero-copy view of a pool slot and is valid ONLY until the next
        ``next()`` (the slot is recycled then); finish any device copy
        within the iteration. When the pool is pinned the view is DMA-able
        (``.cuda(non_blocking=True)`` is a true async copy)."""
        n = len(contents)
        if n == 0:
            return
        first = self.transform(contents[0])
        if isinstance(first, torch.Tensor):
            first = first.numpy()
        first = np.asarray(first)
        chunks = [(s, min(s + self.chunk_size, n))
                  for s in range(0, n, self.chunk_size)]
        if self.workers <= 1:
            for s, e in chunks:
                out = np.empty((e - s,) + first.shape, dtype=first.dtype)
                for i in range(s, e):
                    r = self.transform(contents[i])
                    out[i - s] = r.numpy() if isinstance(r, torch.Tensor) else r
                yield torch.from_numpy(out)
            return
        pool = self._ensure_pool(first)
        free = list(range(pool.slots))
        ready = {}  # seq -> slot
        next_submit = 0
        next_yield = 0
        received = 0
        prev_slot = None
        try:
            while next_yield < len(chunks):
                while free and next_submit < len(chunks):
                    slot = free.pop()
                    s, e = chunks[next_submit]
                    pool.task_q.put((next_submit, slot, list(contents[s:e])))
                    next_submit += 1
                while next_yield not in ready:
                    try:
                        status, seq, slot, payload = pool.res_q.get(timeout=10.0)
                    except queue.Empty:
                        dead = pool.any_dead()
                        if dead:
                            raise RuntimeError(f"decode worker(s) died (pids {dead})")
                        continue
                    received += 1
                    if status == "err":
                        raise RuntimeError(f"decode failed in worker: {payload}")
                    ready[seq] = (slot, payload)
                slot, cnt = ready.pop(next_yield)
                view = torch.from_numpy(pool.view[slot, :cnt])
                if prev_slot is not None:
                    free.append(prev_slot)  # released one step late: the
                prev_slot = slot            # consumer just finished with it
                yield view
                next_yield += 1
        finally:
            # an abandoned/errored iteration leaves results in flight; drain
            # them so the pool can serve the next call with fresh seq ids
            self._drain(pool, next_submit - received)

    def close(self) -> None:
        if self._pool is not None:
            self._pool.close()
            self._pool = None

    def __enter__(self) -> "ParallelDecoder":
        return self

    def __exit__(self, *exc) -> None:
        self.close()
