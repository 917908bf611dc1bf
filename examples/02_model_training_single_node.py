#!/usr/bin/env python3
"""Single-node training from the silver tables.

Equivalent of ``Part 1 .../02_model_training_single_node.py``: load the
train/val tables into memory, build the transfer model (frozen MobileNetV2
base + trainable head), compile Adam + sparse-CE-from-logits + accuracy,
autolog to the tracking store, fit, and log the model.
"""
import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import argparse

import torch

from ddlw_amd.core import setup, tracking
from ddlw_amd.data import read_table
from ddlw_amd.data.preprocess import preprocess_batch
from ddlw_amd.models import build_model
from ddlw_amd.train import Model, autolog

IMG_HEIGHT, IMG_WIDTH, IMG_CHANNELS = 64, 64, 3  # synthetic-tree size
BATCH_SIZE = 32
EPOCHS = 3


def load_split(table: str, img: int):
    tbl = read_table(table, columns=["content", "label_idx"])
    contents = tbl.column("content").to_pylist()
    labels = tbl.column("label_idx").to_pylist()
    xs = preprocess_batch(contents, img, img)
    ys = torch.tensor(labels, dtype=torch.long)
    return xs, ys


def batches(xs, ys, bs):
    return [(xs[i : i + bs], ys[i : i + bs]) for i in range(0, len(xs), bs)]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--root", default=None)
    ap.add_argument("--epochs", type=int, default=EPOCHS)
    ap.add_argument("--img", type=int, default=IMG_HEIGHT)
    args = ap.parse_args()
    setup(root=args.root)

    xs, ys = load_split("silver_train", args.img)
    vx, vy = load_split("silver_val", args.img)
    num_classes = int(ys.max()) + 1

    tracking.set_experiment("single_node_training")
    autolog()
    model = Model(build_model(args.img, args.img, IMG_CHANNELS, num_classes))
    model.compile(optimizer="Adam", learning_rate=1e-3, metrics=("accuracy",))
    with tracking.start_run(run_name="single_node") as run:
        hist = model.fit(
            batches(xs, ys, BATCH_SIZE),
            epochs=args.epochs,
            validation_data=batches(vx, vy, BATCH_SIZE),
        )
        print({k: v[-1] for k, v in hist.history.items()})
        print(f"model logged under runs:/{run.run_id}/model")


if __name__ == "__main__":
    main()
