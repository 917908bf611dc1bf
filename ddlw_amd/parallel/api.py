"""Horovod-equivalent data-parallel API on torch.distributed (RCCL on ROCm).

Reference call sites this re-implements:

- C1 ``hvd.init()``          -> :func:`init` — env rendezvous, one process per
  GPU (``Part 1 .../03_model_training_distributed.py:283``);
- C2 ``hvd.DistributedOptimizer``-> :class:`DistributedOptimizer` — bucketed
  gradient all-reduce overlapped with backward (:302);
- C3 ``BroadcastGlobalVariablesCallback`` -> :func:`broadcast_parameters` /
  :func:`broadcast_optimizer_state` (:308);
- C4 ``MetricAverageCallback``   -> :func:`allreduce_metrics` (:313);
- C6 ``hvd.rank()/local_rank()/size()`` -> same names (:295, :301, ...).

MI355X-native design: the 8-GPU xGMI fabric is a full mesh
of 7 point-to-point links per GPU (~153 GB/s each). RCCL's all-reduce uses
them all when messages are large enough, so gradient fusion **buckets default
to 32 MiB** — each of the 7 per-peer shards is then >4 MiB, keeping every
link bandwidth-bound instead of latency-bound. Collectives run on RCCL's own
internal stream and overlap with the remaining backward compute; ``step()``
waits on the outstanding works before applying the update.

Degenerate world (size 1 / no env rendezvous) runs without a process group —
the reference's ``HorovodRunner(np=-1)`` in-process smoke mode (:385-394).

Data-plane decision (round 2): torch's ProcessGroupNCCL — which IS RCCL on
ROCm — is the one multi-GPU data plane. A round-1 ctypes RCCL wrapper was
removed: it duplicated the same librccl calls behind a layer that could
never be validated multi-rank on the 1-GPU dev lease, and the parts that
matter for xGMI performance (bucket fusion, launch-during-backward overlap,
bucket sizing for the 7-link mesh) live HERE, above the collective call.
"""
from __future__ import annotations

import os
from typing import Dict, Iterable, List, Optional, Tuple

import torch
import torch.distributed as dist

_initialized = False
_world_size = 1
_rank = 0
_local_rank = 0


def init(backend: Optional[str] = None, timeout_s: float = 300.0) -> None:
    """Initialise from the standard env rendezvous (RANK / WORLD_SIZE /
    MASTER_ADDR / MASTER_PORT / LOCAL_RANK). Without env vars: world of 1,
    no process group (in-process mode)."""
    global _initialized, _world_size, _rank, _local_rank
    if _initialized:
        return
    world = int(os.environ.get("WORLD_SIZE", "1"))
    _rank = int(os.environ.get("RANK", "0"))
    _local_rank = int(os.environ.get("LOCAL_RANK", str(_rank)))
    _world_size = world
    if world > 1:
        if backend is None:
            backend = os.environ.get("DDLW_BACKEND")
            if backend is None:
                ngpu = torch.cuda.device_count() if torch.cuda.is_available() else 0
                # RCCL refuses two ranks on one physical GPU — oversubscribed
                # worlds (e.g. np=2 smoke runs on a 1-GPU box) ride gloo,
                # which moves CUDA tensors via host staging
                backend = "nccl" if ngpu >= world else "gloo"
        if torch.cuda.is_available():
            torch.cuda.set_device(_local_rank % torch.cuda.device_count())
        import datetime

        dist.init_process_group(
            backend=backend,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    _initialized = True


def shutdown() -> None:
    global _initialized, _world_size, _rank, _local_rank
    if _initialized and dist.is_initialized():
        dist.destroy_process_group()
    _initialized = False
    _world_size, _rank, _local_rank = 1, 0, 0


def is_initialized() -> bool:
    return _initialized


def rank() -> int:
    return _rank


def local_rank() -> int:
    return _local_rank


def size() -> int:
    return _world_size


def _pg_active() -> bool:
    return _world_size > 1 and dist.is_initialized()


def barrier() -> None:
    if _pg_active():
        dist.barrier()


def allreduce_(tensor: torch.Tensor, average: bool = True) -> torch.Tensor:
    """In-place sum (or mean) all-reduce; identity in a world of 1."""
    if _pg_active():
        dist.all_reduce(tensor, op=dist.ReduceOp.SUM)
        if average:
            tensor.div_(_world_size)
    return tensor


def allreduce_metrics(metrics: Dict[str, float]) -> Dict[str, float]:
    """Average scalar metrics across ranks (MetricAverageCallback, C4)."""
    if not _pg_active() or not metrics:
        return dict(metrics)
    keys = sorted(metrics.keys())
    device = torch.device("cuda", torch.cuda.current_device()) if (
        dist.get_backend() == "nccl"
    ) else torch.device("cpu")
    t = torch.tensor([float(metrics[k]) for k in keys], dtype=torch.float64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    t /= _world_size
    return {k: v for k, v in zip(keys, t.cpu().tolist())}


def broadcast_parameters(module_or_params, root_rank: int = 0,
                         bucket_cap_mb: float = 64.0) -> None:
    """Broadcast model parameters (+ buffers) from root (C3). Ensures every
    rank starts from rank-0's init or restored checkpoint
    (reference comment ``.../03_model_training_distributed.py:305-307``).

    Tensors are coalesced into per-(dtype, device) flat buckets — one
    collective per bucket (~#dtypes total) instead of one per tensor
    (ResNet-50 has ~320 state tensors; Horovod's fusion exists for exactly
    this). CPU-resident buffers ride a single staged device round-trip per
    bucket when the backend is RCCL."""
    if not _pg_active():
        return
    if isinstance(module_or_params, torch.nn.Module):
        tensors: Iterable[torch.Tensor] = list(module_or_params.state_dict().values())
    else:
        tensors = list(module_or_params)
    tensors = [
        t for t in tensors
        if t.dtype.is_floating_point or t.dtype in (torch.int64, torch.int32, torch.uint8)
    ]
    use_nccl = dist.get_backend() == "nccl"
    cap = int(bucket_cap_mb * 1024 * 1024)
    groups: Dict[Tuple[torch.dtype, str], List[torch.Tensor]] = {}
    for t in tensors:
        groups.setdefault((t.dtype, t.device.type), []).append(t)
    for (dtype, devtype), ts in groups.items():
        start = 0
        while start < len(ts):
            nbytes, end = 0, start
            while end < len(ts) and (
                end == start
                or nbytes + ts[end].numel() * ts[end].element_size() <= cap
            ):
                nbytes += ts[end].numel() * ts[end].element_size()
                end += 1
            chunk = ts[start:end]
            start = end
            flat = torch._utils._flatten_dense_tensors([c.detach() for c in chunk])
            if use_nccl and devtype == "cpu":
                # RCCL moves device tensors only: one staged round-trip
                d = flat.to(torch.device("cuda", torch.cuda.current_device()))
                dist.broadcast(d, src=root_rank)
                flat = d.cpu()
            else:
                dist.broadcast(flat, src=root_rank)
            with torch.no_grad():
                for t, piece in zip(
                    chunk, torch._utils._unflatten_dense_tensors(flat, chunk)
                ):
                    t.copy_(piece)  # identity on root


def broadcast_optimizer_state(optimizer: torch.optim.Optimizer, root_rank: int = 0) -> None:
    if not _pg_active():
        return
    tensors = [
        v
        for group in optimizer.param_groups
        for p in group["params"]
        for v in optimizer.state.get(p, {}).values()
        if torch.is_tensor(v)
    ]
    broadcast_parameters(tensors, root_rank=root_rank)


# --------------------------------------------------------------------------- #
# DistributedOptimizer
# --------------------------------------------------------------------------- #


class _Bucket:
    __slots__ = ("params", "bytes", "flat", "work", "ready", "live")

    def __init__(self):
        self.params: List[torch.nn.Parameter] = []
        self.bytes = 0
        self.flat: Optional[torch.Tensor] = None
        self.work = None
        self.ready = 0
        self.live: List[torch.nn.Parameter] = []  # params with grads this step


class DistributedOptimizer:
    """Wraps a torch optimizer with bucketed gradient all-reduce (C2).

    Gradients are coalesced into fusion buckets (default 32 MiB — sized for
    the 7-link xGMI mesh) in reverse parameter order (the
    approximate backward completion order) and all-reduced asynchronously as
    soon as every gradient in a bucket has been produced, overlapping
    communication with the rest of backward. ``step()`` waits for the works,
    averages, and applies the inner optimizer.
    """

    def __init__(
        self,
        optimizer: torch.optim.Optimizer,
        bucket_cap_mb: float = 32.0,
        average: bool = True,
    ):
        from ..utils.trace import get_tracer

        self.optimizer = optimizer
        self.average = average
        self._tracer = get_tracer()  # DDLW_TIMELINE collective events
        self._params: List[torch.nn.Parameter] = [
            p for g in optimizer.param_groups for p in g["params"] if p.requires_grad
        ]
        self._buckets: List[_Bucket] = []
        self._param_bucket: Dict[int, Tuple[_Bucket, int]] = {}
        self._hooks = []
        cap = int(bucket_cap_mb * 1024 * 1024)
        # reverse order ~ backward completion order (last layers first);
        # buckets are per-dtype (bf16 weights + fp32 BN params coexist and
        # flat buffers must be homogeneous)
        open_buckets: Dict[torch.dtype, _Bucket] = {}
        for p in reversed(self._params):
            nbytes = p.numel() * p.element_size()
            bucket = open_buckets.get(p.dtype)
            if bucket is None:
                bucket = open_buckets[p.dtype] = _Bucket()
            if bucket.params and bucket.bytes + nbytes > cap:
                self._buckets.append(bucket)
                bucket = open_buckets[p.dtype] = _Bucket()
            bucket.params.append(p)
            bucket.bytes += nbytes
        for bucket in open_buckets.values():
            if bucket.params:
                self._buckets.append(bucket)
        for b in self._buckets:
            for p in b.params:
                self._param_bucket[id(p)] = (b, 0)
                if _pg_active():
                    h = p.register_post_accumulate_grad_hook(self._make_hook(b))
                    self._hooks.append(h)

    def _make_hook(self, bucket: _Bucket):
        def hook(param: torch.nn.Parameter) -> None:
            bucket.ready += 1
            if bucket.ready == len(bucket.params):
                self._launch(bucket)

        return hook

    def _launch(self, bucket: _Bucket) -> None:
        # params consistently unused across ranks (no grad) stay out of the
        # flat buffer; ranks must agree on graph structure (DDP contract)
        bucket.live = [p for p in bucket.params if p.grad is not None]
        if not bucket.live:
            return
        grads = [p.grad for p in bucket.live]
        flat = torch._utils._flatten_dense_tensors(grads)
        if self._tracer.enabled:
            import time as _t

            self._tracer.event(
                f"allreduce[{flat.numel() * flat.element_size() >> 20}MiB]",
                "collective", _t.time() * 1e6, 1.0, tid=_rank,
            )
        # async: RCCL enqueues on its own comm stream (overlaps backward);
        # the work handle orders the optimizer behind the collective
        bucket.work = dist.all_reduce(flat, op=dist.ReduceOp.SUM, async_op=True)
        bucket.flat = flat

    # -- torch optimizer surface ---------------------------------------- #
    @property
    def param_groups(self):
        return self.optimizer.param_groups

    @property
    def state(self):
        return self.optimizer.state

    def state_dict(self):
        return self.optimizer.state_dict()

    def load_state_dict(self, sd):
        return self.optimizer.load_state_dict(sd)

    def zero_grad(self, set_to_none: bool = True) -> None:
        for b in self._buckets:
            b.ready = 0
            b.flat = None
        self.optimizer.zero_grad(set_to_none=set_to_none)

    def synchronize(self) -> None:
        """Finish outstanding reductions and unflatten back into p.grad."""
        if not _pg_active():
            return
        for b in self._buckets:
            if b.flat is None and any(p.grad is not None for p in b.params):
                # hook missed (grads produced outside autograd, or a bucket
                # containing unused params never hit its ready count)
                self._launch(b)
        for b in self._buckets:
            if b.flat is None:
                continue
            if b.work is not None:
                b.work.wait()
                b.work = None
            if self.average:
                b.flat.div_(_world_size)
            grads = [p.grad for p in b.live]
            for p, g in zip(
                b.live, torch._utils._unflatten_dense_tensors(b.flat, grads)
            ):
                p.grad.copy_(g)
            b.flat = None
            b.ready = 0

    def step(self, closure=None):
        self.synchronize()
        return self.optimizer.step(closure)
