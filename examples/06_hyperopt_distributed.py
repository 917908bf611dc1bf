#!/usr/bin/env python3
"""HPO over nested *distributed* trials.

Equivalent of ``Part 2 .../02_hyperopt_distributed_model.py``: each trial is
itself a data-parallel job (Runner(np=N) inside the objective), so trials
run *sequentially* on the driver with the default Trials — the reference
documents that SparkTrials is incompatible with nested distributed jobs
(P2/02:342-344). Rank 0 of each trial checkpoints per epoch into a per-trial
directory named by the hyperparameters (checkpoint-{epoch}.ckpt) and logs a
nested child run under the parent via the env-carried parent run id.
"""
import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import argparse
import math
import os
import time

import torch

from ddlw_amd.core import setup, tracking
from ddlw_amd.data import make_converter, table_path
from ddlw_amd.models import build_model
from ddlw_amd.parallel import Runner, api
from ddlw_amd.train import (
    BroadcastGlobalVariablesCallback,
    MetricAverageCallback,
    Model,
    ModelCheckpoint,
)
from ddlw_amd.tune import STATUS_OK, Trials, fmin, hp, tpe

IMG = 64
HVD_NUM_PROCESSES = 2


def train_and_evaluate_hvd(learning_rate, dropout, batch_size, checkpoint_dir,
                           root, num_classes, epochs=2):
    """Parameterized distributed train fn (reference P2/02:161-262)."""
    setup(root=root)
    device = None
    if torch.cuda.is_available():
        device = torch.device("cuda", api.local_rank() % torch.cuda.device_count())
        torch.cuda.set_device(device)
    torch.manual_seed(42 + api.rank())
    module = build_model(IMG, IMG, 3, num_classes, dropout=dropout)
    if device is not None:
        module = module.to(device)
    model = Model(module, device=device)
    model.compile(optimizer="Adam", learning_rate=learning_rate * api.size())
    from ddlw_amd.parallel.api import DistributedOptimizer

    model.optimizer = DistributedOptimizer(model.optimizer)

    conv_train = make_converter(str(table_path("silver_train")))
    conv_val = make_converter(str(table_path("silver_val")))
    steps = max(1, len(conv_train) // (batch_size * api.size()))
    val_steps = max(1, len(conv_val) // (batch_size * api.size()))

    callbacks = [BroadcastGlobalVariablesCallback(0), MetricAverageCallback()]
    if api.rank() == 0:
        # per-trial checkpoint dir named by the params (P2/02:206-211)
        trial_dir = os.path.join(
            checkpoint_dir, f"lr_{learning_rate}_dropout_{dropout}_bs_{batch_size}"
        )
        callbacks.append(
            ModelCheckpoint(os.path.join(trial_dir, "checkpoint-{epoch}.ckpt"))
        )
    with conv_train.make_torch_dataset(
        batch_size=batch_size, cur_shard=api.rank(), shard_count=api.size(),
        img_height=IMG, img_width=IMG, device=device,
    ) as train_ds, conv_val.make_torch_dataset(
        batch_size=batch_size, cur_shard=api.rank(), shard_count=api.size(),
        img_height=IMG, img_width=IMG, device=device,
    ) as val_ds:
        hist = model.fit(
            train_ds, steps_per_epoch=steps, epochs=epochs,
            validation_data=val_ds, validation_steps=val_steps,
            callbacks=callbacks, verbose=0,
        )
    val_loss = hist.history.get("val_loss", [0.0])[-1]
    val_acc = hist.history.get("val_accuracy", [0.0])[-1]
    if api.rank() == 0:
        # nested child run under the driver's parent run (P2/02:241-260)
        run = tracking.start_run(
            run_name=f"lr{learning_rate:.2e}_do{dropout:.2f}_bs{batch_size}",
            nested=True,
        )
        run.log_params({"learning_rate": learning_rate, "dropout": dropout,
                        "batch_size": batch_size, "checkpoint_dir": checkpoint_dir})
        run.log_metrics({"val_loss": val_loss, "val_accuracy": val_acc,
                         "accuracy": hist.history.get("accuracy", [0.0])[-1]})
        tracking.end_run()
    return val_loss, val_acc


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--root", default=None)
    ap.add_argument("--max-evals", type=int, default=4)
    ap.add_argument("--np", type=int, default=HVD_NUM_PROCESSES)
    args = ap.parse_args()
    s = setup(root=args.root)
    import json

    label_map = json.loads((table_path("silver_train").parent / "label_to_idx.json").read_text())
    num_classes = len(label_map)
    checkpoint_dir = str(s.root / "checkpoints" / time.strftime("%Y%m%d-%H%M%S"))

    tracking.set_experiment("hyperopt_distributed")
    parent = tracking.start_run(run_name="hpo_distributed_parent")
    os.environ["DDLW_PARENT_RUN_ID"] = parent.run_id

    def objective_function(params):
        bs = int(params["batch_size"])
        val_loss, _ = Runner(np=args.np).run(
            train_and_evaluate_hvd,
            learning_rate=params["learning_rate"],
            dropout=params["dropout"],
            batch_size=bs,
            checkpoint_dir=checkpoint_dir,
            root=str(s.root),
            num_classes=num_classes,
        )
        return {"loss": val_loss, "status": STATUS_OK}

    search_space = {
        "learning_rate": hp.loguniform("learning_rate", math.log(1e-4), math.log(1e-1)),
        "dropout": hp.uniform("dropout", 0.2, 0.8),
        "batch_size": hp.choice("batch_size", [16, 32, 64]),
    }
    # sequential Trials on the driver: nested distributed trials cannot be
    # task-parallelized (reference P2/02:342-344)
    best = fmin(objective_function, search_space, algo=tpe.suggest,
                max_evals=args.max_evals, trials=Trials())
    tracking.end_run()
    print("best (hp.choice values are indices):", best)

    import pathlib

    ckpts = sorted(pathlib.Path(checkpoint_dir).rglob("checkpoint-*.ckpt"))
    print(f"{len(ckpts)} checkpoints under {checkpoint_dir}")
    df = tracking.search_runs(
        filter_string=f'tags.mlflow.parentRunId = "{parent.run_id}"',
        order_by=["metrics.val_accuracy DESC"],
    )
    print("best child run:", df.iloc[0]["run_id"], "val_acc:", df.iloc[0]["metrics.val_accuracy"])


if __name__ == "__main__":
    main()
