"""Fused SGD optimizer — one multi-tensor kernel per step (north-star
"fused SGD step"; replaces torch.optim.SGD's per-tensor foreach ops).

Momentum + weight decay. Two parameter classes:
- fp32 params: updated in place (grad fp32);
- bf16 params (conv/linear weights kept in bf16 so no per-step autocast
  casts): the optimizer holds an fp32 MASTER copy + momentum; the kernel
  reads the bf16 grad, updates the master, and writes the bf16 shadow —
  all in the same pass (mixed-precision SGD exactly like the reference's
  LossScale-free bf16 recipe).

Momentum buffers start at zero, so v = mu*v + g reproduces torch.optim.SGD's
buf-initialized-to-grad first step. The chunk descriptor table is built once
per group and refreshed only if grad storage moves, so the steady-state step
is one kernel launch per group.
"""
from __future__ import annotations

from collections import defaultdict
from copy import deepcopy
from typing import Dict, List, Tuple

import torch

from . import binding


def _load_state_dict_no_cast(opt: torch.optim.Optimizer, state_dict) -> None:
    """Load optimizer state WITHOUT torch's param-dtype casting.

    torch.optim.Optimizer.load_state_dict casts floating-point state tensors
    to the param's dtype — for bf16 params that turns the fp32 master /
    momentum / exp_avg buffers into bf16, but the fused kernels read those
    buffers through raw fp32 pointers (out-of-bounds reads + precision loss).
    This loader maps saved state to params positionally and only moves
    tensors to the param's device, preserving dtype exactly.
    """
    sd = deepcopy(state_dict)
    groups = opt.param_groups
    saved_groups = sd["param_groups"]
    if len(groups) != len(saved_groups):
        raise ValueError("loaded state dict has a different number of parameter groups")
    id_map: Dict[int, torch.Tensor] = {}
    for g, sg in zip(groups, saved_groups):
        if len(g["params"]) != len(sg["params"]):
            raise ValueError("loaded state dict contains a parameter group that "
                             "doesn't match the size of optimizer's group")
        id_map.update(zip(sg["params"], g["params"]))
    new_state: Dict[torch.Tensor, dict] = {}
    for k, v in sd["state"].items():
        p = id_map[k]
        new_state[p] = {
            kk: (vv.to(p.device) if isinstance(vv, torch.Tensor) else vv)
            for kk, vv in v.items()
        }
    opt.state = defaultdict(dict, new_state)
    for g, sg in zip(groups, saved_groups):
        for kk, vv in sg.items():
            if kk != "params":
                g[kk] = vv


class FusedSGD(torch.optim.Optimizer):
    def __init__(self, params, lr: float = 0.1, momentum: float = 0.9,
                 weight_decay: float = 0.0):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._desc: Dict[int, Tuple[tuple, torch.Tensor, int]] = {}

    def _state_for(self, p: torch.Tensor):
        st = self.state[p]
        if "momentum_buffer" not in st:
            if p.dtype == torch.bfloat16:
                st["master"] = p.detach().float()
                st["momentum_buffer"] = torch.zeros_like(st["master"])
            else:
                st["momentum_buffer"] = torch.zeros_like(p)
        return st

    def _group_desc(self, gi: int, params: List[torch.Tensor]):
        key = tuple((p.data_ptr(), p.grad.data_ptr(), p.numel()) for p in params)
        cached = self._desc.get(gi)
        if cached is not None and cached[0] == key:
            return cached[1], cached[2]
        rows = []
        max_numel = 0
        for p in params:
            st = self._state_for(p)
            if p.dtype == torch.bfloat16:
                rows.append(
                    (st["master"].data_ptr(), p.grad.data_ptr(),
                     st["momentum_buffer"].data_ptr(), p.data_ptr(), p.numel(),
                     1 if p.grad.dtype == torch.bfloat16 else 0)
                )
            else:
                assert p.grad.dtype == torch.float32, p.grad.dtype
                rows.append(
                    (p.data_ptr(), p.grad.data_ptr(),
                     st["momentum_buffer"].data_ptr(), 0, p.numel(), 0)
                )
            max_numel = max(max_numel, p.numel())
        cpu = torch.tensor(rows, dtype=torch.int64).pin_memory()
        desc = cpu.to(params[0].device, non_blocking=True)
        # keep the pinned source alive until at least the next rebuild (the
        # async H2D may still be reading it when this function returns)
        self._desc[gi] = (key, desc, max_numel, cpu)
        return desc, max_numel

    def load_state_dict(self, state_dict):
        # no-cast load (fp32 master/momentum of bf16 params must stay fp32
        # for the raw-pointer fused kernel); also invalidate the descriptor
        # cache — it holds raw device pointers into the replaced state
        _load_state_dict_no_cast(self, state_dict)
        self._desc.clear()

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for gi, group in enumerate(self.param_groups):
            params = [p for p in group["params"] if p.grad is not None]
            if not params:
                continue
            if not params[0].is_cuda:
                # CPU fallback = plain SGD-with-momentum semantics (oracle)
                for p in params:
                    st = self._state_for(p)
                    if p.dtype == torch.bfloat16:
                        master = st["master"]
                        g = p.grad.float() + group["weight_decay"] * master
                        st["momentum_buffer"].mul_(group["momentum"]).add_(g)
                        master.add_(st["momentum_buffer"], alpha=-group["lr"])
                        p.copy_(master.to(torch.bfloat16))
                    else:
                        g = p.grad + group["weight_decay"] * p
                        st["momentum_buffer"].mul_(group["momentum"]).add_(g)
                        p.add_(st["momentum_buffer"], alpha=-group["lr"])
                continue
            desc, max_numel = self._group_desc(gi, params)
            binding.fused_sgd(
                desc, len(params), max_numel, group["lr"], group["momentum"],
                group["weight_decay"], False,
            )
        return loss


class FusedAdam(torch.optim.Optimizer):
    """Fused Adam (kernel K10 — the reference's optimizer, Adam 1e-3 at
    P1/02:201): one multi-tensor kernel per step; fp32 m/v (+ fp32 master for
    bf16 params with the bf16 shadow updated in the same pass)."""

    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._desc: Dict[int, Tuple[tuple, torch.Tensor, int]] = {}
        self._step_t = 0

    def _state_for(self, p: torch.Tensor):
        st = self.state[p]
        if "exp_avg" not in st:
            if p.dtype == torch.bfloat16:
                st["master"] = p.detach().float()
                base = st["master"]
            else:
                base = p
            st["exp_avg"] = torch.zeros_like(base)
            st["exp_avg_sq"] = torch.zeros_like(base)
        return st

    def _group_desc(self, gi: int, params: List[torch.Tensor]):
        key = tuple((p.data_ptr(), p.grad.data_ptr(), p.numel()) for p in params)
        cached = self._desc.get(gi)
        if cached is not None and cached[0] == key:
            return cached[1], cached[2]
        rows = []
        max_numel = 0
        for p in params:
            st = self._state_for(p)
            if p.dtype == torch.bfloat16:
                rows.append(
                    (st["master"].data_ptr(), p.grad.data_ptr(),
                     st["exp_avg"].data_ptr(), st["exp_avg_sq"].data_ptr(),
                     p.data_ptr(), p.numel(),
                     1 if p.grad.dtype == torch.bfloat16 else 0)
                )
            else:
                rows.append(
                    (p.data_ptr(), p.grad.data_ptr(), st["exp_avg"].data_ptr(),
                     st["exp_avg_sq"].data_ptr(), 0, p.numel(), 0)
                )
            max_numel = max(max_numel, p.numel())
        cpu = torch.tensor(rows, dtype=torch.int64).pin_memory()
        desc = cpu.to(params[0].device, non_blocking=True)
        # keep the pinned source alive until at least the next rebuild (the
        # async H2D may still be reading it when this function returns)
        self._desc[gi] = (key, desc, max_numel, cpu)
        return desc, max_numel

    def state_dict(self):
        # persist the shared step count per-param under 'step' so a fresh
        # instance resumes bias correction at the right t (copy the inner
        # dicts — super() returns references into live state)
        sd = super().state_dict()
        sd["state"] = {k: dict(v) for k, v in sd["state"].items()}
        for st in sd["state"].values():
            st["step"] = self._step_t
        return sd

    def load_state_dict(self, state_dict):
        # no-cast load (fp32 master/m/v of bf16 params must stay fp32 for
        # the raw-pointer fused kernel); restore the step count; invalidate
        # the descriptor cache holding raw pointers into the replaced state
        _load_state_dict_no_cast(self, state_dict)
        self._step_t = max(
            (st.get("step", 0) for st in self.state.values()), default=0)
        for st in self.state.values():
            st.pop("step", None)
        self._desc.clear()

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        self._step_t += 1
        t = self._step_t
        for gi, group in enumerate(self.param_groups):
            params = [p for p in group["params"] if p.grad is not None]
            if not params:
                continue
            b1, b2 = group["betas"]
            if not params[0].is_cuda:
                for p in params:
                    st = self._state_for(p)
                    master = st.get("master")
                    tgt = master if master is not None else p
                    g = p.grad.float() + group["weight_decay"] * tgt
                    st["exp_avg"].mul_(b1).add_(g, alpha=1 - b1)
                    st["exp_avg_sq"].mul_(b2).addcmul_(g, g, value=1 - b2)
                    mh = st["exp_avg"] / (1 - b1 ** t)
                    vh = st["exp_avg_sq"] / (1 - b2 ** t)
                    tgt.addcdiv_(mh, vh.sqrt().add_(group["eps"]), value=-group["lr"])
                    if master is not None:
                        p.copy_(master.to(torch.bfloat16))
                continue
            desc, max_numel = self._group_desc(gi, params)
            binding.fused_adam(
                desc, len(params), max_numel, group["lr"], b1, b2,
                group["eps"], group["weight_decay"],
                1.0 / (1.0 - b1 ** t), 1.0 / (1.0 - b2 ** t),
            )
        return loss
