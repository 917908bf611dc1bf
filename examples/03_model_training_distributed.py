#!/usr/bin/env python3
"""Distributed data-parallel training — the core reference flow.

Equivalent of ``Part 1 .../03_model_training_distributed.py``: converters
over the train/val tables, a ``train_and_evaluate`` function run first with
``Runner(np=-1)`` (in-process smoke mode) and then ``Runner(np=N)``
(multi-process, RCCL on GPU / gloo on CPU), with the full Horovod-style
callback set: broadcast, metric averaging, LR warmup, plateau decay; rank 0
re-attaches to the driver's tracking run and logs the model.
"""
import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import argparse
import os

import torch

from ddlw_amd.core import setup, tracking
from ddlw_amd.core.model_io import load_model
from ddlw_amd.data import make_converter, table_path
from ddlw_amd.models import build_model
from ddlw_amd.parallel import Runner, api
from ddlw_amd.train import (
    BroadcastGlobalVariablesCallback,
    LearningRateWarmupCallback,
    MetricAverageCallback,
    Model,
    ReduceLROnPlateau,
)

BATCH_SIZE = 32
EPOCHS = 3
IMG = 64


def train_and_evaluate(root: str, run_id: str, img: int, epochs: int,
                       num_classes: int, batch_size: int = BATCH_SIZE):
    """Runs on every rank (the reference's train_and_evaluate_hvd,
    P1/03:282-375)."""
    setup(root=root)
    os.environ.setdefault("DDLW_TRACKING_URI", str(setup().tracking_uri))

    device = None
    if torch.cuda.is_available():
        device = torch.device("cuda", api.local_rank() % torch.cuda.device_count())
        torch.cuda.set_device(device)

    torch.manual_seed(42 + api.rank())
    module = build_model(img, img, 3, num_classes)
    if device is not None:
        module = module.to(device).to(memory_format=torch.channels_last)
    model = Model(module, device=device)
    # LR scaled by world size (P1/03:301)
    model.compile(optimizer="Adam", learning_rate=1e-3 * api.size())
    from ddlw_amd.parallel.api import DistributedOptimizer

    model.optimizer = DistributedOptimizer(model.optimizer)

    conv_train = make_converter(str(table_path("silver_train")))
    conv_val = make_converter(str(table_path("silver_val")))
    steps = max(1, len(conv_train) // (batch_size * api.size()))
    val_steps = max(1, len(conv_val) // (batch_size * api.size()))

    callbacks = [
        BroadcastGlobalVariablesCallback(0),
        MetricAverageCallback(),
        LearningRateWarmupCallback(warmup_epochs=min(5, epochs)),
        ReduceLROnPlateau(patience=10),
    ]
    with conv_train.make_torch_dataset(
        batch_size=batch_size, cur_shard=api.rank(), shard_count=api.size(),
        img_height=img, img_width=img, device=device,
    ) as train_ds, conv_val.make_torch_dataset(
        batch_size=batch_size, cur_shard=api.rank(), shard_count=api.size(),
        img_height=img, img_width=img, device=device,
    ) as val_ds:
        hist = model.fit(
            train_ds,
            steps_per_epoch=steps,
            epochs=epochs,
            validation_data=val_ds,
            validation_steps=val_steps,
            callbacks=callbacks,
            verbose=1 if api.rank() == 0 else 0,
        )

    if api.rank() == 0 and run_id:
        run = tracking.start_run(run_id=run_id)  # re-attach (P1/03:361-373)
        run.log_params({"world_size": api.size(), "batch_size": batch_size})
        run.log_metrics({k: v[-1] for k, v in hist.history.items()})
        from ddlw_amd.core.model_io import log_model

        log_model(model.module, "model")
        tracking.end_run()
    return (
        hist.history.get("val_loss", [0.0])[-1],
        hist.history.get("val_accuracy", [0.0])[-1],
    )


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--np", type=int, default=2)
    ap.add_argument("--root", default=None)
    ap.add_argument("--epochs", type=int, default=EPOCHS)
    args = ap.parse_args()
    s = setup(root=args.root)
    import json

    label_map = json.loads((table_path("silver_train").parent / "label_to_idx.json").read_text())
    num_classes = len(label_map)

    tracking.set_experiment("distributed_training")

    # smoke mode first: np=-1 runs in-process (P1/03:385-394)
    print("== smoke run (np=-1) ==")
    Runner(np=-1).run(
        train_and_evaluate, root=str(s.root), run_id="", img=IMG, epochs=1,
        num_classes=num_classes,
    )

    print(f"== distributed run (np={args.np}) ==")
    run = tracking.start_run(run_name="distributed")
    run_id = run.run_id
    tracking.end_run()
    val_loss, val_acc = Runner(np=args.np).run(
        train_and_evaluate, root=str(s.root), run_id=run_id, img=IMG,
        epochs=args.epochs, num_classes=num_classes,
    )
    print(f"val_loss={val_loss:.4f} val_acc={val_acc:.4f}")
    model = load_model(f"runs:/{run_id}/model")
    print("reloaded:", type(model).__name__)


if __name__ == "__main__":
    main()
