#!/usr/bin/env python3
"""Packaged-model (pyfunc) training + distributed batch inference.

Equivalent of ``Part 2 .../03_pyfunc_distributed_inference.py``:
- ``FlowerPyFunc`` packaged predict-function: load_context reads
  ``img_params_dict.json`` + the trained model from packaged artifacts;
  predict = PIL decode/resize -> model -> argmax -> class name (incl. the
  str->bytes content workaround);
- ``train_model_petastorm_data_ingest`` equivalent: converter-fed training
  that logs the img-params dict, the torch model, and the pyfunc bundle;
- single-node predict on a few rows, then the predict-UDF fanned out across
  local GPU workers on 1000 rows (bounded by table size here).
"""
import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import argparse
import json
from dataclasses import dataclass, field

import numpy as np
import torch

from ddlw_amd.core import DataCfg, setup, tracking
from ddlw_amd.core.model_io import load_model as load_torch_model, log_model
from ddlw_amd.data import make_converter, read_table, table_path
from ddlw_amd.infer import PythonModel, load_model, log_model as log_pyfunc, predict_udf
from ddlw_amd.models import build_model
from ddlw_amd.train import EarlyStopping, Model

IMG = 64
BATCH_SIZE = 128


class FlowerPyFunc(PythonModel):
    """Reference P2/03:157-234."""

    CLASSES = None  # resolved from img_params at load

    def load_context(self, context):
        with open(context.artifacts["img_params_dict_path"]) as f:
            self.img_params = json.load(f)
        self.classes = self.img_params["classes"]
        self.model = load_torch_model(context.artifacts["torch_model_path"])
        self.model.eval()

    def predict(self, context, model_input):
        from ddlw_amd.data.preprocess import preprocess_pil

        h, w = self.img_params["img_height"], self.img_params["img_width"]
        arrs = np.stack([preprocess_pil(c, h, w) for c in model_input])
        x = torch.from_numpy(arrs).permute(0, 3, 1, 2).float()
        if torch.cuda.is_available():
            self.model.cuda()
            x = x.cuda()
        outs = []
        with torch.no_grad():
            for i in range(0, len(x), BATCH_SIZE):
                outs.append(self.model(x[i : i + BATCH_SIZE]).cpu())
        idx = torch.cat(outs).argmax(-1).numpy()
        return np.take(self.classes, idx)


def train_model_with_converter(data_cfg: DataCfg, epochs: int = 2):
    """Reference P2/03:253-377 (train + package)."""
    label_map = json.loads(
        (table_path("silver_train").parent / "label_to_idx.json").read_text()
    )
    classes = [c for c, _ in sorted(label_map.items(), key=lambda kv: kv[1])]
    with tracking.start_run(run_name="pyfunc_model_training") as run:
        img_params = {
            "img_height": data_cfg.img_height,
            "img_width": data_cfg.img_width,
            "img_channels": data_cfg.img_channels,
            "num_classes": len(classes),
            "classes": classes,
        }
        run.log_dict(img_params, "img_params_dict.json")

        device = torch.device("cuda:0") if torch.cuda.is_available() else None
        module = build_model(data_cfg.img_height, data_cfg.img_width,
                             data_cfg.img_channels, len(classes))
        if device:
            module = module.to(device)
        model = Model(module, device=device)
        model.compile(optimizer="Adam", learning_rate=1e-3)

        conv_train = make_converter(str(table_path(data_cfg.train_table)))
        conv_val = make_converter(str(table_path(data_cfg.val_table)))
        steps = max(1, len(conv_train) // data_cfg.batch_size)
        with conv_train.make_torch_dataset(
            batch_size=data_cfg.batch_size, img_height=data_cfg.img_height,
            img_width=data_cfg.img_width, device=device,
        ) as tds, conv_val.make_torch_dataset(
            batch_size=data_cfg.batch_size, img_height=data_cfg.img_height,
            img_width=data_cfg.img_width, device=device,
        ) as vds:
            model.fit(
                tds, steps_per_epoch=steps, epochs=epochs,
                validation_data=vds, validation_steps=1,
                callbacks=[EarlyStopping(monitor="val_loss", min_delta=1e-2, patience=3)],
                verbose=1,
            )
            val = model.evaluate(vds, steps=2, return_dict=True)
        run.log_metrics({f"val_{k}": v for k, v in val.items()})
        model_uri = log_model(model.module, "model")
        pyfunc_uri = log_pyfunc(
            "pyfunc_model",
            FlowerPyFunc(),
            artifacts={
                "img_params_dict_path": f"runs:/{run.run_id}/img_params_dict.json",
                "torch_model_path": model_uri,
            },
        )
    return pyfunc_uri


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--root", default=None)
    ap.add_argument("--workers", type=int, default=4)
    args = ap.parse_args()
    setup(root=args.root)
    tracking.set_experiment("pyfunc_inference")
    cfg = DataCfg(img_height=IMG, img_width=IMG, batch_size=32)
    uri = train_model_with_converter(cfg)
    print("pyfunc:", uri)

    rows = read_table("silver", columns=["content"]).column("content").to_pylist()
    # single-node smoke on 10 rows (P2/03:446-450)
    m = load_model(uri)
    preds10 = m.predict(rows[:10])
    print("single-node:", list(preds10))

    # distributed fan-out on up to 1000 rows (P2/03:466-472)
    with predict_udf(uri, num_workers=args.workers) as udf:
        preds = udf(rows[:1000])
    print(f"fanned out {len(preds)} predictions; first 5: {preds[:5]}")
    assert list(preds10) == [str(p) for p in preds[:10]]


if __name__ == "__main__":
    main()
