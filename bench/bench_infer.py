#!/usr/bin/env python3
"""BASELINE config 5: packaged-model batch inference fan-out — the pyfunc
predict-UDF fanned to N local GPU workers (the 03_pyfunc path).

Trains nothing: packages a random-init ResNet-50 pyfunc, then measures
images/sec over a synthetic JPEG batch fanned across workers (decode on CPU
in each worker, batched bf16 model forward on its GPU).
"""
import argparse
import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np  # noqa: E402
import torch  # noqa: E402

from ddlw_amd.core import setup, tracking  # noqa: E402
from ddlw_amd.core.model_io import log_model  # noqa: E402
from ddlw_amd.infer import PythonModel, load_model, log_model as log_pyfunc, predict_udf  # noqa: E402
from ddlw_amd.models import build_resnet50  # noqa: E402


class ResNetPyFunc(PythonModel):
    def load_context(self, context):
        import json as _json

        from ddlw_amd.core.model_io import load_model as load_torch

        with open(context.artifacts["img_params"]) as f:
            self.params = _json.load(f)
        self.model = load_torch(context.artifacts["model"])
        self.model.eval()
        if torch.cuda.is_available():
            self.model = self.model.cuda().to(memory_format=torch.channels_last)
            # bf16 conv/fc weights ONCE (BN stats stay fp32 — the eval-BN
            # fold reads them); avoids per-batch autocast weight casts
            for m in self.model.modules():
                if isinstance(m, (torch.nn.Conv2d, torch.nn.Linear)):
                    m.to(torch.bfloat16)

    def predict(self, context, model_input):
        import functools

        from ddlw_amd.data.decode import ParallelDecoder, decode_resize_u8

        h = self.params["img_height"]
        bs = self.params["batch_size"]
        use_cuda = torch.cuda.is_available()
        if not hasattr(self, "_decoder"):
            # forked process decode pool (each UDF worker owns its own);
            # uint8 HWC out — the [-1,1] normalize runs on-GPU, fused.
            # chunk = model batch so imap yields model-ready batches that
            # decode ahead while the GPU runs the previous one; the slot
            # ring is pinned -> .cuda(non_blocking) DMAs straight from it
            self._decoder = ParallelDecoder(
                functools.partial(decode_resize_u8, img_height=h, img_width=h),
                workers=max(2, (os.cpu_count() or 8)
                            // max(1, torch.cuda.device_count() or 1)),
                chunk_size=bs,
                pin=use_cuda,
            )
        outs = []
        rows = list(model_input)
        with torch.no_grad():
            for u8 in self._decoder.imap(rows):
                n_real = u8.shape[0]
                if use_cuda:
                    from ddlw_amd.ops import normalize_u8_bf16

                    d = u8.cuda(non_blocking=True)
                    if n_real < bs:  # pad: a new batch shape would trigger
                        pad = d[-1:].expand(bs - n_real, *d.shape[1:])
                        d = torch.cat([d, pad])  # a fresh MIOpen find
                    x = normalize_u8_bf16(d.permute(0, 3, 1, 2))
                    logits = self.model(x)
                else:
                    x = u8.permute(0, 3, 1, 2).float() / 127.5 - 1.0
                    logits = self.model(x)
                outs.append(logits.float().argmax(-1).cpu()[:n_real])
        return torch.cat(outs).numpy().astype(str)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=None)
    ap.add_argument("--rows", type=int, default=1000)
    ap.add_argument("--batch-size", type=int, default=128)
    args = ap.parse_args()
    n = args.gpus or max(1, torch.cuda.device_count())

    setup()
    tracking.set_experiment("bench_infer")
    with tracking.start_run(run_name="bench_pyfunc") as run:
        m = build_resnet50(num_classes=1000)
        model_uri = log_model(m, "model")
        run.log_dict({"img_height": 224, "batch_size": args.batch_size}, "img_params.json")
        uri = log_pyfunc(
            "pyfunc_model",
            ResNetPyFunc(),
            artifacts={
                "img_params": f"runs:/{run.run_id}/img_params.json",
                "model": model_uri,
            },
        )

    from ddlw_amd.data.synthetic import make_synthetic_dataset

    contents, _ = make_synthetic_dataset(args.rows, 224, 224, jpeg=True, seed=1)
    udf = predict_udf(uri, num_workers=n, gpus=list(range(n)) if torch.cuda.is_available() else [])
    # warm fan-out (worker model load + MIOpen) then timed
    udf(contents[: args.batch_size])
    t0 = time.perf_counter()
    preds = udf(contents)
    wall = time.perf_counter() - t0
    udf.close()
    assert len(preds) == args.rows
    print(
        json.dumps(
            {
                "metric": "images/sec (whole node) ResNet-50 pyfunc batch inference",
                "value": round(args.rows / wall, 2),
                "unit": "images/sec",
                "n_gpus": n if torch.cuda.is_available() else 0,
                "steps": 1,
                "warmup": 1,
                "ms_per_step": round(wall * 1000, 1),
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": None,
                "dtype": "bf16" if torch.cuda.is_available() else "fp32",
                "data": "synthetic-jpeg",
                "config": {"model": "resnet50", "rows": args.rows,
                           "batch_size": args.batch_size,
                           "parallelism": f"udf-fanout-{n}"},
            }
        )
    )


if __name__ == "__main__":
    main()
