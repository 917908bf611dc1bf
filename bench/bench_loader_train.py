#!/usr/bin/env python3
"""BASELINE config 2: ResNet-50 bf16 on one MI355X fed by the Parquet shard
loader (the Petastorm-equivalent path) with pinned H2D staging on a side
stream — i.e. the flagship training step of ``bench.py`` but with real
JPEG-decode -> resize -> batch -> pinned-H2D data flow instead of a
pre-staged synthetic pool.

Reference path: ``Part 1 .../03_model_training_distributed.py:197-234``
(converter.make_torch_dataset -> train loop). Data here is synthetic JPEGs
(no network for TF-flowers): 224px, quality-85, 5 classes, re-decoded every
epoch by a CPU worker pool, so the number includes the full input pipeline.

    python bench/bench_loader_train.py --steps 30 --warmup 5
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pyarrow as pa
import torch


def _decode224(content) -> torch.Tensor:
    """JPEG bytes -> uint8 HWC tensor (resize only if needed); the [-1,1]
    normalize runs on-device via the fused normalize_u8_bf16 kernel."""
    import io

    from PIL import Image

    img = Image.open(io.BytesIO(content)).convert("RGB")
    if img.size != (224, 224):
        img = img.resize((224, 224), Image.BILINEAR)
    return torch.from_numpy(np.asarray(img).copy())


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch-size", type=int, default=256)
    p.add_argument("--rows", type=int, default=4096)
    p.add_argument("--workers", type=int, default=min(os.cpu_count() or 8, 64))
    p.add_argument("--loader-only", action="store_true",
                   help="time the input pipeline alone (no model): upper "
                        "bound on loader-fed throughput")
    args = p.parse_args()

    from ddlw_amd.core.config import setup
    from ddlw_amd.data.loader import make_converter
    from ddlw_amd.data.synthetic import make_synthetic_dataset
    from ddlw_amd.models import build_resnet50
    from ddlw_amd.ops import FusedSGD, normalize_u8_bf16, require_lib, softmax_cross_entropy

    assert torch.cuda.is_available(), "config-2 bench needs a GPU"
    require_lib()
    device = torch.device("cuda", 0)
    torch.cuda.set_device(device)

    setup(root="/tmp/ddlw_bench_loader")
    print(f"[bench_loader] building {args.rows}-row synthetic JPEG parquet ...", flush=True)
    contents, labels = make_synthetic_dataset(args.rows, 224, 224, jpeg=True, num_classes=5)
    table = pa.table({"content": pa.array(contents, pa.binary()), "label_idx": labels})
    conv = make_converter(table, row_group_rows=64)

    torch.manual_seed(1234)
    model = build_resnet50(num_classes=5).to(device).to(memory_format=torch.channels_last)
    for m in model.modules():
        if isinstance(m, (torch.nn.Conv2d, torch.nn.Linear)):
            m.to(torch.bfloat16)
    opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9, weight_decay=1e-4)
    model.train()

    steps, warmup, bs = args.steps, args.warmup, args.batch_size
    with conv.make_torch_dataset(
        batch_size=bs,
        cur_shard=0,
        shard_count=1,
        num_epochs=None,  # infinite cycling, Petastorm default
        workers_count=args.workers,
        device=device,
        transform=_decode224,
        prefetch=4,
    ) as loader:
        it = iter(loader)

        if args.loader_only:
            for _ in range(warmup):
                next(it)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(steps):
                next(it)
            torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            it.close()
            conv.delete()
            print(json.dumps({
                "metric": "loader images/sec (no model)",
                "value": round(bs * steps / dt, 2),
                "ms_per_batch": round(dt / steps * 1000, 3),
                "workers": args.workers,
            }), flush=True)
            return

        def step_fn():
            imgs, labs = next(it)  # uint8 NHWC on device, staged via side stream
            x = imgs.permute(0, 3, 1, 2)  # NCHW view == channels_last memory
            opt.zero_grad(set_to_none=True)
            loss = softmax_cross_entropy(model(normalize_u8_bf16(x)), labs)
            loss.backward()
            opt.step()

        for _ in range(warmup):
            step_fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(steps):
            step_fn()
        torch.cuda.synchronize()
        elapsed = time.perf_counter() - t0
        it.close()  # stop the producer thread before teardown

    conv.delete()
    torch.cuda.synchronize()
    out = {
        "metric": "images/sec (1 GPU) ResNet-50 224px, Parquet-loader-fed",
        "value": round(bs * steps / elapsed, 2),
        "unit": "images/sec",
        "n_gpus": 1,
        "steps": steps,
        "warmup": warmup,
        "ms_per_step": round(elapsed / steps * 1000, 3),
        "higher_is_better": True,
        "dtype": "bf16",
        "data": "synthetic-jpeg-parquet",
        "config": {
            "model": "resnet50",
            "global_batch": bs,
            "image_size": 224,
            "parallelism": "dp1",
            "loader_workers": args.workers,
            "rows": args.rows,
        },
    }
    print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
