"""``ddlw_amd.parallel`` — data parallelism on RCCL over xGMI.

The ``hvd``-style surface the reference exercises:
``init / rank / local_rank / size / DistributedOptimizer /
broadcast_parameters / allreduce_metrics`` plus the local multi-process
launcher ``Runner`` (HorovodRunner contract, including ``np=-1``)."""

from .api import (
    init,
    shutdown,
    is_initialized,
    rank,
    local_rank,
    size,
    allreduce_,
    allreduce_metrics,
    broadcast_parameters,
    broadcast_optimizer_state,
    barrier,
    DistributedOptimizer,
)
from .runner import Runner

__all__ = [
    "init",
    "shutdown",
    "is_initialized",
    "rank",
    "local_rank",
    "size",
    "allreduce_",
    "allreduce_metrics",
    "broadcast_parameters",
    "broadcast_optimizer_state",
    "barrier",
    "DistributedOptimizer",
    "Runner",
]
