#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 @ 224px training throughput (images/sec,
whole node) — the BASELINE.json north-star metric.

Usage (driver contract):
    python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this under torch.distributed.run with one rank
per GPU (RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* in env) over RCCL.

What a "step" is: one full data-parallel training step of ResNet-50 (bf16
compute, fp32 master weights) on a fresh synthetic batch staged host->device
through pinned memory on a side stream: H2D copy + normalize (the
preprocess_input [-1,1] transform) + forward + backward + bucketed gradient
all-reduce (N>1) + SGD momentum update. Per-GPU batch 256 (the reference's
distributed batch size, ``Part 1 .../03_model_training_distributed.py:81``);
weak scaling (global batch = 256*N).

Timing: W untimed warmup steps, then exactly K steps bracketed by
barrier + torch.cuda.synchronize on both sides; MAX elapsed over ranks;
rank 0 prints ONE JSON line.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch-size", type=int, default=256, help="per-GPU batch")
    p.add_argument("--model", type=str, default="resnet50")
    p.add_argument("--dtype", type=str, default="bf16")
    p.add_argument("--no-hip-ops", action="store_true", help="disable ddlw HIP kernels (stock-op baseline)")
    p.add_argument("--graph", action="store_true", default=None, help="capture the step in a hipGraph")
    p.add_argument("--no-graph", dest="graph", action="store_false")
    return p.parse_args()


def main() -> None:
    args = parse_args()
    from ddlw_amd.parallel import api

    api.init()
    world = api.size()
    rank = api.rank()
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        device = torch.device("cuda", api.local_rank() % torch.cuda.device_count())
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    batch = args.batch_size if use_cuda else 8
    steps, warmup = args.steps, args.warmup
    num_classes = 1000

    hip_ops = use_cuda and not args.no_hip_ops
    if use_cuda and args.no_hip_ops:
        os.environ["DDLW_DISABLE_HIP_OPS"] = "1"
    torch.manual_seed(1234)
    from ddlw_amd.models import build_resnet50
    from ddlw_amd.ops import FusedSGD, normalize_u8_bf16, require_lib, softmax_cross_entropy

    if hip_ops:
        require_lib()  # fail loudly if the HIP extension is missing on GPU

    model = build_resnet50(num_classes=num_classes).to(device)
    if use_cuda:
        model = model.to(memory_format=torch.channels_last)
    if hip_ops:
        # bf16 weights for convs/fc (no per-step autocast casts); BN params
        # and stats stay fp32; FusedSGD keeps fp32 masters
        for m in model.modules():
            if isinstance(m, (torch.nn.Conv2d, torch.nn.Linear)):
                m.to(torch.bfloat16)

    if hip_ops:
        base_opt = FusedSGD(model.parameters(), lr=0.1 * world, momentum=0.9, weight_decay=1e-4)
    else:
        base_opt = torch.optim.SGD(model.parameters(), lr=0.1 * world, momentum=0.9, weight_decay=1e-4)
    if world > 1:
        opt = api.DistributedOptimizer(base_opt, bucket_cap_mb=32.0)
        api.broadcast_parameters(model, root_rank=0)
    else:
        opt = base_opt

    # synthetic data: rotating pool of pinned uint8 host batches (channels_last)
    # staged H2D on a side stream (the loader's staging path), normalized to
    # bf16 [-1,1] on device (the preprocess_input transform, fused kernel)
    g = torch.Generator().manual_seed(4321 + rank)
    pool = 4
    host_batches = [
        torch.randint(0, 256, (batch, 3, 224, 224), dtype=torch.uint8, generator=g)
        .contiguous(memory_format=torch.channels_last)
        for _ in range(pool)
    ]
    labels_pool = [
        torch.randint(0, num_classes, (batch,), dtype=torch.long, generator=g)
        for _ in range(pool)
    ]
    if use_cuda:
        host_batches = [b.pin_memory() for b in host_batches]
        labels_pool = [l.pin_memory() for l in labels_pool]
        side = torch.cuda.Stream(device)
        amp_dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32

    loss_fn = torch.nn.CrossEntropyLoss()

    def step_fn(i: int) -> None:
        hb, hl = host_batches[i % pool], labels_pool[i % pool]
        if use_cuda:
            ev = torch.cuda.Event()
            with torch.cuda.stream(side):
                d_img = hb.to(device, non_blocking=True)
                d_lab = hl.to(device, non_blocking=True)
                ev.record(side)
            torch.cuda.current_stream().wait_event(ev)
            opt.zero_grad(set_to_none=True)
            if hip_ops:
                x = normalize_u8_bf16(d_img)
                logits = model(x)
                loss = softmax_cross_entropy(logits, d_lab)
            else:
                x = d_img.to(amp_dtype).mul_(1.0 / 127.5).sub_(1.0)
                with torch.autocast("cuda", dtype=amp_dtype):
                    logits = model(x)
                    loss = loss_fn(logits.float(), d_lab)
            loss.backward()
            opt.step()
        else:
            x = hb.float().mul_(1.0 / 127.5).sub_(1.0)
            opt.zero_grad(set_to_none=True)
            loss = loss_fn(model(x), hl)
            loss.backward()
            opt.step()

    model.train()

    # hipGraph capture: H2D staging stays OUTSIDE the graphs (fresh data per
    # step into static device buffers); normalize+fwd+bwd+opt are captured
    # once per ping-pong buffer and replayed — removes launch gaps for the
    # ~600-kernel step, and the copy for step i+1 overlaps replay of step i.
    # default OFF: measured ~4% slower than eager on this step (the ~600
    # kernels are large; launch gaps are already hidden) — use --graph to force
    use_graph = args.graph if args.graph is not None else False
    if use_cuda and use_graph:
        bufs_img = [
            torch.empty(host_batches[0].shape, dtype=torch.uint8, device=device).contiguous(
                memory_format=torch.channels_last
            )
            for _ in range(2)
        ]
        bufs_lab = [torch.empty(batch, dtype=torch.long, device=device) for _ in range(2)]

        def compute_from(d_img, d_lab):
            opt.zero_grad(set_to_none=False)
            if hip_ops:
                x = normalize_u8_bf16(d_img)
                logits = model(x)
                loss = softmax_cross_entropy(logits, d_lab)
            else:
                x = d_img.to(amp_dtype).mul(1.0 / 127.5).sub(1.0)
                with torch.autocast("cuda", dtype=amp_dtype):
                    logits = model(x)
                    loss = loss_fn(logits.float(), d_lab)
            loss.backward()
            opt.step()
            return loss

        # eager warmup first (MIOpen find / autotune must finish pre-capture)
        for i in range(max(warmup, 3)):
            s = i % 2
            bufs_img[s].copy_(host_batches[i % pool], non_blocking=True)
            bufs_lab[s].copy_(labels_pool[i % pool], non_blocking=True)
            compute_from(bufs_img[s], bufs_lab[s])
        torch.cuda.synchronize()
        graphs = []
        for s in range(2):
            gr = torch.cuda.CUDAGraph()
            with torch.cuda.graph(gr):
                compute_from(bufs_img[s], bufs_lab[s])
            graphs.append(gr)
        torch.cuda.synchronize()

        main = torch.cuda.current_stream()
        ready = [torch.cuda.Event(), torch.cuda.Event()]
        free = [torch.cuda.Event(), torch.cuda.Event()]
        for e in free:
            e.record(main)

        def _prefetch(i: int) -> None:
            s = i % 2
            side.wait_event(free[s])  # prior replay reading buf s must finish
            with torch.cuda.stream(side):
                bufs_img[s].copy_(host_batches[i % pool], non_blocking=True)
                bufs_lab[s].copy_(labels_pool[i % pool], non_blocking=True)
                ready[s].record(side)

        state = {"next": None}

        def graph_step(i: int) -> None:
            if state["next"] != i:
                torch.cuda.synchronize()
                _prefetch(i)
            s = i % 2
            main.wait_event(ready[s])
            graphs[s].replay()
            free[s].record(main)
            _prefetch(i + 1)
            state["next"] = i + 1

        step_fn = graph_step  # noqa: F811

    for i in range(warmup):
        step_fn(i)

    if use_cuda:
        torch.cuda.synchronize()
    api.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(steps):
        step_fn(warmup + i)
    if use_cuda:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    api.barrier()

    # MAX over ranks
    t = torch.tensor([elapsed], dtype=torch.float64)
    if world > 1:
        import torch.distributed as dist

        if dist.get_backend() == "nccl":
            t = t.to(device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.cpu().item())

    total_images = batch * world * steps
    ips = total_images / elapsed
    if rank == 0:
        out = {
            "metric": "images/sec (whole node) ResNet-50 224px",
            "value": round(ips, 2),
            "unit": "images/sec",
            "n_gpus": world,
            "steps": steps,
            "warmup": warmup,
            "ms_per_step": round(elapsed / steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype if use_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": "resnet50" if use_cuda else "resnet50-cpu-fallback",
                "global_batch": batch * world,
                "per_gpu_batch": batch,
                "seq_len": None,
                "image_size": 224,
                "parallelism": f"dp{world}",
                "hip_ops": bool(use_cuda and not args.no_hip_ops),
            },
        }
        print(json.dumps(out), flush=True)
    api.shutdown()


if __name__ == "__main__":
    main()
