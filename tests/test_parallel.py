"""Distributed tests on gloo, world_size 2, CPU.

Covers: Runner np=-1 in-process mode, multi-process allreduce/broadcast
correctness, DistributedOptimizer gradient averaging vs a single-process
oracle, metric averaging.
"""
import os

import pytest
import torch

from ddlw_amd.parallel import Runner, api


def test_np_minus_one_in_process():
    def fn(a, b):
        from ddlw_amd.parallel import api

        assert api.size() == 1 and api.rank() == 0
        return a + b

    assert Runner(np=-1).run(fn, a=2, b=3) == 5


def _allreduce_worker():
    from ddlw_amd.parallel import api

    t = torch.tensor([float(api.rank() + 1)])
    api.allreduce_(t, average=True)  # (1+2)/2 = 1.5
    metrics = api.allreduce_metrics({"loss": float(api.rank())})  # (0+1)/2
    assert abs(metrics["loss"] - 0.5) < 1e-6
    return float(t.item())


def test_runner_two_proc_allreduce():
    out = Runner(np=2, timeout_s=120).run(_allreduce_worker)
    assert abs(out - 1.5) < 1e-6


def _broadcast_worker():
    from ddlw_amd.parallel import api

    m = torch.nn.Linear(4, 2)
    with torch.no_grad():
        m.weight.fill_(float(api.rank() + 1))
        m.bias.fill_(float(api.rank() + 1))
    api.broadcast_parameters(m, root_rank=0)
    # every rank must now hold rank-0's weights (== 1.0)
    assert torch.all(m.weight == 1.0) and torch.all(m.bias == 1.0)
    return True


def test_runner_broadcast_parameters():
    assert Runner(np=2, timeout_s=120).run(_broadcast_worker)


def _distopt_worker(seed):
    from ddlw_amd.parallel import api

    torch.manual_seed(seed)
    m = torch.nn.Linear(8, 4)
    api.broadcast_parameters(m, root_rank=0)
    opt = api.DistributedOptimizer(torch.optim.SGD(m.parameters(), lr=0.1), bucket_cap_mb=0.0001)
    # per-rank different batch
    g = torch.Generator().manual_seed(100 + api.rank())
    x = torch.randn(4, 8, generator=g)
    y = m(x).sum()
    opt.zero_grad()
    y.backward()
    opt.step()
    return {k: v.detach().clone() for k, v in m.state_dict().items()}


def test_distributed_optimizer_matches_oracle():
    out = Runner(np=2, timeout_s=120).run(_distopt_worker, seed=7)

    # single-process oracle: average the two batches' gradients
    torch.manual_seed(7)
    m = torch.nn.Linear(8, 4)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    xs = [torch.randn(4, 8, generator=torch.Generator().manual_seed(100 + r)) for r in range(2)]
    loss = sum(m(x).sum() for x in xs) / 2
    opt.zero_grad()
    loss.backward()
    opt.step()
    for k, v in m.state_dict().items():
        assert torch.allclose(out[k], v, atol=1e-6), k


def _failure_worker():
    from ddlw_amd.parallel import api

    if api.rank() == 1:
        raise RuntimeError("boom")
    import torch.distributed as dist

    # rank 0 would block on a collective; give it something short
    return "rank0-done"


def test_runner_worker_failure_detected():
    with pytest.raises(RuntimeError):
        Runner(np=2, timeout_s=60).run(_failure_worker)


def _distopt_fused_worker(seed):
    """DistributedOptimizer wrapping FusedSGD (the bench.py N>1 config),
    CPU-fallback semantics vs a single-process oracle."""
    from ddlw_amd.ops.optim import FusedSGD
    from ddlw_amd.parallel import api

    torch.manual_seed(seed)
    m = torch.nn.Linear(8, 4)
    api.broadcast_parameters(m, root_rank=0)
    opt = api.DistributedOptimizer(
        FusedSGD(m.parameters(), lr=0.1, momentum=0.9), bucket_cap_mb=0.0001
    )
    g = torch.Generator().manual_seed(300 + api.rank())
    x = torch.randn(4, 8, generator=g)
    for _ in range(2):
        opt.zero_grad()
        m(x).sum().backward()
        opt.step()
    return {k: v.detach().clone() for k, v in m.state_dict().items()}


def test_distributed_fused_sgd_matches_oracle():
    out = Runner(np=2, timeout_s=120).run(_distopt_fused_worker, seed=11)

    torch.manual_seed(11)
    m = torch.nn.Linear(8, 4)
    opt = torch.optim.SGD(m.parameters(), lr=0.1, momentum=0.9)
    xs = [torch.randn(4, 8, generator=torch.Generator().manual_seed(300 + r)) for r in range(2)]
    for _ in range(2):
        opt.zero_grad()
        (sum(m(x).sum() for x in xs) / 2).backward()
        opt.step()
    for k, v in m.state_dict().items():
        assert torch.allclose(out[k], v, atol=1e-5), k


def _elastic_worker(ckpt_dir):
    """Crashes rank 1 on the first gang attempt after 'epoch 0' completes;
    on restart, resumes from the checkpoint file (SURVEY.md §5.3 epoch-
    granular restart)."""
    attempt = int(os.environ["DDLW_RESTART_ATTEMPT"])
    ckpt = os.path.join(ckpt_dir, "epoch.txt")
    start_epoch = int(open(ckpt).read()) + 1 if os.path.exists(ckpt) else 0
    done = start_epoch
    for epoch in range(start_epoch, 3):
        api.barrier()
        if api.rank() == 0:  # rank-0-only checkpoint write (reference layout)
            with open(ckpt, "w") as f:
                f.write(str(epoch))
        api.barrier()
        done = epoch + 1
        if attempt == 0 and api.rank() == 1 and epoch == 0:
            os._exit(17)  # simulated hard worker death mid-job
    return {"attempt": attempt, "resumed_from": start_epoch, "epochs_done": done}


def test_runner_gang_restart_resumes_from_checkpoint(tmp_path):
    out = Runner(np=2, timeout_s=120, max_restarts=1).run(
        _elastic_worker, ckpt_dir=str(tmp_path)
    )
    assert out["attempt"] == 1          # first gang died, second succeeded
    assert out["resumed_from"] == 1     # epoch 0's checkpoint was picked up
    assert out["epochs_done"] == 3


def test_runner_no_restart_still_aborts(tmp_path):
    with pytest.raises(RuntimeError, match="exited with code"):
        Runner(np=2, timeout_s=60, max_restarts=0).run(
            _elastic_worker, ckpt_dir=str(tmp_path)
        )


def _hang_worker():
    import time as _t

    if api.rank() == 1:
        _t.sleep(60)
    return True


def test_runner_timeout_aborts_hung_gang():
    import time as _t

    t0 = _t.time()
    with pytest.raises((TimeoutError, RuntimeError)):
        Runner(np=2, timeout_s=6).run(_hang_worker)
    assert _t.time() - t0 < 45  # aborted well before the 60s hang


def _unused_param_worker(seed):
    """Model with a parameter that never receives a gradient (unused head):
    the bucketed optimizer must reduce the live grads and leave the unused
    param's grad None — without crashing or silently skipping buckets."""
    torch.manual_seed(seed)
    lin = torch.nn.Linear(8, 4)
    unused = torch.nn.Parameter(torch.randn(3))
    params = [unused, lin.weight, lin.bias]
    opt = api.DistributedOptimizer(
        torch.optim.SGD(params, lr=0.1), bucket_cap_mb=0.001
    )
    api.broadcast_parameters(params)
    x = torch.randn(4, 8) + api.rank()
    lin(x).sum().backward()
    opt.step()
    return {
        "unused_grad_none": unused.grad is None,
        "w": lin.weight.detach().clone(),
    }


def test_distributed_optimizer_with_unused_param():
    out = Runner(np=2, timeout_s=120).run(_unused_param_worker, seed=3)
    assert out["unused_grad_none"]
    assert torch.isfinite(out["w"]).all()


def _coalesced_broadcast_worker(seed):
    """broadcast must coalesce into per-(dtype,device) flat
    buckets — collective count ~ #dtype groups, not #tensors — and still
    leave every rank with rank-0's exact state."""
    import torch.distributed as dist

    from ddlw_amd.parallel import api

    torch.manual_seed(seed + api.rank())  # divergent init across ranks
    m = torch.nn.Sequential(
        torch.nn.Conv2d(3, 8, 3), torch.nn.BatchNorm2d(8), torch.nn.Linear(8, 4)
    )
    m[2].weight.data = m[2].weight.data.to(torch.bfloat16)  # second dtype
    n_tensors = len(list(m.state_dict().values()))
    calls = []
    orig = dist.broadcast

    def counting(t, *a, **k):
        calls.append(t.numel())
        return orig(t, *a, **k)

    dist.broadcast = counting
    try:
        api.broadcast_parameters(m, root_rank=0)
    finally:
        dist.broadcast = orig
    assert len(calls) <= 4, f"{len(calls)} collectives for {n_tensors} tensors"
    assert len(calls) < n_tensors
    import hashlib

    h = hashlib.sha256()
    for k, v in m.state_dict().items():
        h.update(k.encode())
        h.update(v.float().numpy().tobytes())
    return h.hexdigest()


def test_broadcast_parameters_coalesced():
    # run twice with different seeds; rank-0 digest returned — both ranks
    # asserted equal inside via the returned digest equality across runs
    d1 = Runner(np=2, timeout_s=120).run(_coalesced_broadcast_worker, seed=10)
    d2 = Runner(np=2, timeout_s=120).run(_coalesced_broadcast_worker, seed=10)
    assert d1 == d2  # deterministic given rank-0 seed


def _coalesced_broadcast_parity_worker(seed):
    """All ranks must hold identical state after the coalesced broadcast."""
    from ddlw_amd.parallel import api

    torch.manual_seed(seed + api.rank())
    m = torch.nn.Sequential(torch.nn.Linear(6, 6), torch.nn.BatchNorm1d(6))
    m[0].weight.data = m[0].weight.data.to(torch.bfloat16)
    api.broadcast_parameters(m, root_rank=0)
    sd = {k: v.clone() for k, v in m.state_dict().items()}
    # gather every rank's digest on rank 0 via all_reduce of a checksum
    import hashlib

    h = hashlib.sha256()
    for k in sorted(sd):
        h.update(sd[k].float().numpy().tobytes())
    digest = int.from_bytes(h.digest()[:6], "big")
    t = torch.tensor([float(digest)], dtype=torch.float64)
    mx, mn = t.clone(), t.clone()
    import torch.distributed as dist

    dist.all_reduce(mx, op=dist.ReduceOp.MAX)
    dist.all_reduce(mn, op=dist.ReduceOp.MIN)
    assert torch.equal(mx, mn), "ranks diverge after broadcast"
    return True


def test_broadcast_parameters_rank_parity():
    assert Runner(np=2, timeout_s=120).run(_coalesced_broadcast_parity_worker, seed=3)
