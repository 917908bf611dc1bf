#!/usr/bin/env python3
"""BASELINE config 4: HPO throughput — N parallel single-GPU ResNet-50
trials (the 02_hyperopt path's task parallelism), one trial per GPU via
LocalTrials/HIP_VISIBLE_DEVICES pinning.

Reports aggregate images/sec across all concurrent trials.
"""
import argparse
import json
import math
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def trial(params):
    import torch

    from ddlw_amd.models import build_resnet50
    from ddlw_amd.ops import FusedSGD, require_lib, softmax_cross_entropy

    steps = int(params["steps"])
    batch = int(params["batch"])
    device = torch.device("cuda:0")
    require_lib()
    torch.manual_seed(0)
    model = build_resnet50(num_classes=1000).to(device).to(memory_format=torch.channels_last)
    for m in model.modules():
        if isinstance(m, (torch.nn.Conv2d, torch.nn.Linear)):
            m.to(torch.bfloat16)
    opt = FusedSGD(model.parameters(), lr=params["lr"], momentum=0.9)
    x = torch.randn(batch, 3, 224, 224, device=device).to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last
    )
    y = torch.randint(0, 1000, (batch,), device=device)
    model.train()

    def one():
        opt.zero_grad(set_to_none=True)
        logits = model(x)
        loss = softmax_cross_entropy(logits, y)
        loss.backward()
        opt.step()

    for _ in range(3):
        one()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        one()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return {"loss": -batch * steps / dt, "status": "ok", "images_per_sec": batch * steps / dt}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=None)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--batch-size", type=int, default=256)
    args = ap.parse_args()
    import numpy as np
    import torch

    from ddlw_amd.tune import LocalTrials, fmin, hp

    n = args.gpus or max(1, torch.cuda.device_count())
    trials = LocalTrials(parallelism=n, gpus=list(range(n)))
    space = {
        "lr": hp.loguniform("lr", math.log(1e-3), math.log(1e-1)),
        "steps": hp.choice("steps", [args.steps]),
        "batch": hp.choice("batch", [args.batch_size]),
    }
    t0 = time.perf_counter()
    fmin(trial, space, max_evals=n, trials=trials, verbose=False,
         rstate=np.random.default_rng(0))
    wall = time.perf_counter() - t0
    per_trial = [t["result"].get("images_per_sec", 0.0) for t in trials.trials]
    # aggregate steady-state throughput over the concurrently running trials
    # (wall includes per-process startup + MIOpen find, reported separately)
    agg = sum(per_trial)
    print(
        json.dumps(
            {
                "metric": "images/sec (whole node) ResNet-50 HPO trials",
                "value": round(agg, 2),
                "wall_s": round(wall, 2),
                "unit": "images/sec",
                "n_gpus": n,
                "steps": args.steps,
                "warmup": 3,
                "ms_per_step": round(wall / args.steps * 1000, 3),
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": None,
                "dtype": "bf16",
                "data": "synthetic",
                "config": {"model": "resnet50", "global_batch": args.batch_size * n,
                           "parallelism": f"hpo-{n}x1gpu",
                           "per_trial_images_per_sec": [round(v, 1) for v in per_trial]},
            }
        )
    )


if __name__ == "__main__":
    main()
