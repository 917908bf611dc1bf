#!/usr/bin/env python3
"""BASELINE config 1: 3-layer CNN on 64x64 synthetic JPEGs, single-process
CPU (the 02_model_training_single_node path with no GPU).

Measures images/sec through the real pipeline: JPEG decode + preprocess +
fwd/bwd + Adam step on CPU.
"""
import argparse
import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402

from ddlw_amd.data.synthetic import make_synthetic_dataset  # noqa: E402
from ddlw_amd.data.preprocess import preprocess_batch  # noqa: E402
from ddlw_amd.models import build_small_cnn  # noqa: E402
from ddlw_amd.train import Model  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch-size", type=int, default=32)
    args = ap.parse_args()

    contents, labels = make_synthetic_dataset(
        args.batch_size * 4, img_height=64, img_width=64, num_classes=5, jpeg=True
    )
    model = Model(build_small_cnn(64, 64, num_classes=5)).compile("Adam", learning_rate=1e-3)

    def step(i):
        lo = (i % 4) * args.batch_size
        xs = preprocess_batch(contents[lo : lo + args.batch_size], 64, 64)
        ys = torch.tensor(labels[lo : lo + args.batch_size])
        model.train_step(xs, ys)

    for i in range(args.warmup):
        step(i)
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(args.warmup + i)
    dt = time.perf_counter() - t0
    print(
        json.dumps(
            {
                "metric": "images/sec 3-layer CNN 64px CPU",
                "value": round(args.batch_size * args.steps / dt, 2),
                "unit": "images/sec",
                "n_gpus": 0,
                "steps": args.steps,
                "warmup": args.warmup,
                "ms_per_step": round(dt / args.steps * 1000, 3),
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": None,
                "dtype": "fp32",
                "data": "synthetic-jpeg",
                "config": {"model": "small_cnn", "global_batch": args.batch_size,
                           "image_size": 64, "parallelism": "single-process-cpu"},
            }
        )
    )


if __name__ == "__main__":
    main()
