#!/usr/bin/env python3
"""Parallel HPO of single-device trials.

Equivalent of ``Part 2 .../01_hyperopt_single_machine_model.py``: TPE over
{optimizer, lr, dropout}, trials run concurrently by LocalTrials (each
pinned to its own GPU when GPUs exist), child runs under a parent tracking
run, best run found via search_runs, registered and promoted to Production.
"""
import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import argparse
import math

import torch

from ddlw_amd.core import setup, tracking
from ddlw_amd.core.model_io import load_model
from ddlw_amd.data import read_table
from ddlw_amd.data.preprocess import preprocess_batch
from ddlw_amd.models import build_model
from ddlw_amd.train import Model, autolog
from ddlw_amd.tune import LocalTrials, STATUS_OK, fmin, hp, tpe

NUM_CLASSES = 5
BATCH_SIZE = 32
IMG = 64


def objective_function(params):
    """One trial (reference P2/01:133-181): rebuild datasets, train, return
    {'loss': -accuracy, 'status': STATUS_OK}."""
    setup()
    autolog()
    tbl = read_table("silver_train", columns=["content", "label_idx"])
    xs = preprocess_batch(tbl.column("content").to_pylist(), IMG, IMG)
    ys = torch.tensor(tbl.column("label_idx").to_pylist())
    vt = read_table("silver_val", columns=["content", "label_idx"])
    vx = preprocess_batch(vt.column("content").to_pylist(), IMG, IMG)
    vy = torch.tensor(vt.column("label_idx").to_pylist())
    batches = [(xs[i : i + BATCH_SIZE], ys[i : i + BATCH_SIZE]) for i in range(0, len(xs), BATCH_SIZE)]

    device = torch.device("cuda:0") if torch.cuda.is_available() else None
    module = build_model(IMG, IMG, 3, int(ys.max()) + 1, dropout=params["dropout"])
    if device:
        module = module.to(device)
    model = Model(module, device=device)
    model.compile(optimizer=params["optimizer"], learning_rate=params["learning_rate"])
    with tracking.start_run(nested=True):
        model.fit(batches, epochs=2, verbose=0)
        loss, acc = model.evaluate([(vx, vy)])
        tracking.log_metric("accuracy", acc)
    return {"loss": -acc, "status": STATUS_OK}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--root", default=None)
    ap.add_argument("--max-evals", type=int, default=8)
    ap.add_argument("--parallelism", type=int, default=4)
    args = ap.parse_args()
    setup(root=args.root)

    search_space = {
        "optimizer": hp.choice("optimizer", ["Adam", "Adadelta"]),
        "learning_rate": hp.loguniform("learning_rate", math.log(1e-4), math.log(1e-1)),
        "dropout": hp.uniform("dropout", 0.1, 0.9),
    }
    tracking.set_experiment("hyperopt_single_machine")
    with tracking.start_run(run_name="hpo_parent") as parent:
        best = fmin(
            objective_function,
            search_space,
            algo=tpe.suggest,
            max_evals=args.max_evals,
            trials=LocalTrials(parallelism=args.parallelism),
        )
        # NOTE: hp.choice entries come back as the INDEX (hyperopt contract)
        print("best:", best)

    df = tracking.search_runs(
        filter_string=f'tags.mlflow.parentRunId = "{parent.run_id}"',
        order_by=["metrics.accuracy DESC"],
    )
    best_run_id = df.iloc[0]["run_id"]
    print("best child run:", best_run_id, "acc:", df.iloc[0]["metrics.accuracy"])

    uri = f"runs:/{best_run_id}/model"
    mv = tracking.register_model(uri, "flower_classifier")
    tracking.transition_model_version_stage("flower_classifier", mv["version"], "Production")
    m = load_model("models:/flower_classifier/production")
    print("production model loaded:", type(m).__name__)


if __name__ == "__main__":
    main()
