"""Native RCCL communicator layer (parallel/hip/rccl_comm.cpp + parallel.native).

World-of-1 GPU tests: communicator bootstrap, in-place allreduce (ncclAvg —
identity over 1 rank), broadcast identity, side-stream event ordering.
Multi-rank semantics of the same bucket logic are covered on CPU by
``tests/test_parallel.py`` (gloo, world_size 2); RCCL refuses two ranks on
one physical GPU, so the >1-rank native path is exercised only on multi-GPU
nodes (the driver's scaling runs use the default torch.distributed path).
"""
import pytest
import torch

gpu = pytest.mark.gpu


@pytest.fixture(scope="module")
def comm():
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    from ddlw_amd.parallel import native

    assert native.available(), "libddlw_rccl.so missing — build must have failed"
    c = native.NativeComm(nranks=1, rank=0, device=torch.device("cuda", 0))
    yield c
    c.destroy()


@gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16, torch.float64])
def test_allreduce_world1_identity(comm, dtype):
    x = torch.randn(1 << 16, device="cuda", dtype=torch.float32).to(dtype)
    ref = x.clone()
    ev = comm.allreduce_(x, average=True)  # avg over world of 1 == identity
    torch.cuda.current_stream().wait_event(ev)
    torch.cuda.synchronize()
    assert torch.equal(x, ref)


@gpu
def test_allreduce_sum_world1(comm):
    x = torch.full((4096,), 3.0, device="cuda")
    ev = comm.allreduce_(x, average=False)
    torch.cuda.current_stream().wait_event(ev)
    torch.cuda.synchronize()
    assert torch.equal(x, torch.full_like(x, 3.0))


@gpu
def test_broadcast_world1_identity(comm):
    x = torch.randn(8192, device="cuda")
    ref = x.clone()
    comm.broadcast_(x, root=0)
    torch.cuda.synchronize()
    assert torch.equal(x, ref)


@gpu
def test_allreduce_overlaps_side_stream(comm):
    """The collective must be ordered after the producing stream's writes."""
    x = torch.zeros(1 << 20, device="cuda")
    x.add_(7.0)  # enqueued on current stream before the collective
    ev = comm.allreduce_(x, average=True)
    torch.cuda.current_stream().wait_event(ev)
    torch.cuda.synchronize()
    assert torch.equal(x, torch.full_like(x, 7.0))
