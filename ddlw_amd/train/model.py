"""Keras-like ``Model`` facade over a PyTorch-ROCm train loop.

Reference contract (SURVEY.md §2.5 "TF/Keras runtime" row):
``model.compile(optimizer, loss, metrics)`` then
``model.fit(train_ds, steps_per_epoch, epochs, validation_data,
validation_steps, callbacks, verbose)`` -> ``History`` with a ``history``
dict (``Part 1 .../02_model_training_single_node.py:198-215``,
``.../03_model_training_distributed.py:353-370``), and
``model.evaluate(ds, steps)`` -> [loss, accuracy]
(``Part 2 .../01_hyperopt_single_machine_model.py:177``).

``autolog()`` replaces ``mlflow.tensorflow.autolog()``: when enabled, ``fit``
logs params + per-epoch metrics to the active tracking run and the final
model under ``runs:/<id>/model`` (rank 0 only).
"""
from __future__ import annotations

import time
from typing import Dict, Iterable, List, Optional, Sequence, Union

import torch
import torch.nn.functional as F

from ..core import tracking
from ..core.model_io import log_model
from ..utils.trace import ChromeTracer, get_tracer
from .callbacks import Callback, MetricAverageCallback

_autolog_enabled = False


def autolog(enable: bool = True) -> None:
    global _autolog_enabled
    _autolog_enabled = enable


def _make_optimizer(spec, params, lr: Optional[float] = None):
    if isinstance(spec, torch.optim.Optimizer):
        return spec
    from ..parallel.api import DistributedOptimizer

    if isinstance(spec, DistributedOptimizer):
        return spec
    name = str(spec)
    lr = 1e-3 if lr is None else lr
    key = name.lower()
    params = list(params)
    # fused ddlw optimizers on GPU (K10); stock torch otherwise
    on_gpu = bool(params) and params[0].is_cuda
    if on_gpu and key in ("adam", "sgd"):
        from ..ops.optim import FusedAdam, FusedSGD

        if key == "adam":
            return FusedAdam(params, lr=lr)
        return FusedSGD(params, lr=lr, momentum=0.0)
    table = {
        "adam": torch.optim.Adam,
        "adadelta": torch.optim.Adadelta,
        "sgd": torch.optim.SGD,
        "adamw": torch.optim.AdamW,
    }
    if key not in table:
        raise ValueError(f"unknown optimizer {spec!r}")
    return table[key](params, lr=lr)


def sparse_categorical_crossentropy_from_logits(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """The reference's loss (``.../02_model_training_single_node.py:202``);
    fused HIP kernel (K9) on GPU, stock cross_entropy on CPU."""
    from ..ops.layers import softmax_cross_entropy

    return softmax_cross_entropy(logits, labels)


class History:
    def __init__(self):
        self.history: Dict[str, List[float]] = {}

    def _append(self, logs: Dict[str, float]) -> None:
        for k, v in logs.items():
            self.history.setdefault(k, []).append(v)


def _accuracy(logits, labels):
    """K11 on the hip kernel for CUDA logits; torch ops otherwise."""
    if logits.is_cuda and logits.dim() == 2:
        try:
            from ..ops import binding

            return binding.accuracy(logits, labels)
        except Exception:
            pass
    return (logits.argmax(-1) == labels).float().mean()


class Model:
    """Wraps an ``nn.Module`` with compile/fit/evaluate."""

    def __init__(self, module: torch.nn.Module, device: Optional[torch.device] = None):
        self.module = module
        if device is None:
            try:
                device = next(module.parameters()).device
            except StopIteration:
                device = torch.device("cpu")
        self.device = device
        self.optimizer = None
        self.loss_fn = sparse_categorical_crossentropy_from_logits
        self.metrics: Sequence[str] = ()
        self.stop_training = False
        self._steps_per_epoch: Optional[int] = None

    # ------------------------------------------------------------------ #
    def compile(
        self,
        optimizer: Union[str, torch.optim.Optimizer] = "Adam",
        loss=None,
        metrics: Sequence[str] = ("accuracy",),
        learning_rate: Optional[float] = None,
    ) -> "Model":
        trainable = [p for p in self.module.parameters() if p.requires_grad]
        self.optimizer = _make_optimizer(optimizer, trainable, learning_rate)
        if loss is not None:
            self.loss_fn = loss
        self.metrics = tuple(metrics)
        return self

    # ------------------------------------------------------------------ #
    def _move(self, x: torch.Tensor) -> torch.Tensor:
        return x.to(self.device, non_blocking=True) if x.device != self.device else x

    def _forward_loss(self, images: torch.Tensor, labels: torch.Tensor):
        logits = self.module(images)
        loss = self.loss_fn(logits, labels)
        return logits, loss

    def train_step(self, images: torch.Tensor, labels: torch.Tensor) -> Dict[str, float]:
        self.module.train()
        images, labels = self._move(images), self._move(labels)
        self.optimizer.zero_grad(set_to_none=True)
        logits, loss = self._forward_loss(images, labels)
        loss.backward()
        self.optimizer.step()
        out = {"loss": float(loss.detach())}
        if "accuracy" in self.metrics:
            out["accuracy"] = float(_accuracy(logits.detach(), labels))
        return out

    @torch.no_grad()
    def eval_step(self, images: torch.Tensor, labels: torch.Tensor) -> Dict[str, float]:
        self.module.eval()
        images, labels = self._move(images), self._move(labels)
        logits, loss = self._forward_loss(images, labels)
        out = {"loss": float(loss)}
        if "accuracy" in self.metrics:
            out["accuracy"] = float(_accuracy(logits, labels))
        return out

    # ------------------------------------------------------------------ #
    def fit(
        self,
        data: Iterable,
        steps_per_epoch: Optional[int] = None,
        epochs: int = 1,
        validation_data: Optional[Iterable] = None,
        validation_steps: Optional[int] = None,
        callbacks: Sequence[Callback] = (),
        verbose: int = 1,
    ) -> History:
        if self.optimizer is None:
            raise RuntimeError("call compile() before fit()")
        # An infinite stream (Petastorm num_epochs=None semantics) with no
        # steps_per_epoch would never end an epoch — fail loudly instead
        # (Keras shares this footgun; we don't).
        if steps_per_epoch is None and getattr(data, "num_epochs", 0) is None:
            raise ValueError(
                "fit(): dataset advertises infinite epochs (num_epochs=None) "
                "but steps_per_epoch was not given — the first epoch would "
                "never end. Pass steps_per_epoch (e.g. len(converter) // "
                "(batch_size * size()))."
            )
        from ..parallel import api

        self.stop_training = False
        self._steps_per_epoch = steps_per_epoch
        history = History()
        # MetricAverage must run before LR-schedule callbacks (reference
        # comment .../03_model_training_distributed.py:310-313): stable-sort
        # metric-average callbacks to the front.
        callbacks = sorted(
            callbacks, key=lambda c: 0 if isinstance(c, MetricAverageCallback) else 1
        )
        for cb in callbacks:
            cb.set_model(self)
        for cb in callbacks:
            cb.on_train_begin()

        run = tracking.active_run() if _autolog_enabled else None
        if run is not None and api.rank() == 0:
            run.log_params(
                {
                    "optimizer_name": type(self.optimizer).__name__,
                    "learning_rate": self.optimizer.param_groups[0]["lr"],
                    "epochs": epochs,
                    "steps_per_epoch": steps_per_epoch or -1,
                }
            )

        # Horovod-Timeline equivalent: DDLW_TIMELINE=<path> -> chrome trace
        from ..parallel import api as _api

        tracer = get_tracer()
        data_iter = iter(data)
        for epoch in range(epochs):
            for cb in callbacks:
                cb.on_epoch_begin(epoch)
            t0 = time.time()
            agg: Dict[str, float] = {}
            n = 0
            step = 0
            while steps_per_epoch is None or step < steps_per_epoch:
                with tracer.span(f"data[e{epoch}s{step}]", "data", tid=_api.rank()):
                    try:
                        images, labels = next(data_iter)
                    except StopIteration:
                        if steps_per_epoch is None:
                            data_iter = iter(data)  # next epoch restarts iterator
                            break
                        data_iter = iter(data)
                        try:
                            images, labels = next(data_iter)
                        except StopIteration:
                            break
                for cb in callbacks:
                    cb.on_batch_begin(step)
                with tracer.span(f"train_step[e{epoch}s{step}]", "step", tid=_api.rank()):
                    logs = self.train_step(images, labels)
                for cb in callbacks:
                    cb.on_batch_end(step, logs)
                for k, v in logs.items():
                    agg[k] = agg.get(k, 0.0) + v
                n += 1
                step += 1
            epoch_logs = {k: v / max(n, 1) for k, v in agg.items()}

            if validation_data is not None:
                val_logs = self.evaluate(
                    validation_data, steps=validation_steps, return_dict=True
                )
                epoch_logs.update({f"val_{k}": v for k, v in val_logs.items()})

            for cb in callbacks:
                cb.on_epoch_end(epoch, epoch_logs)
            history._append(epoch_logs)
            if run is not None and api.rank() == 0:
                run.log_metrics(epoch_logs, step=epoch)
            if verbose and api.rank() == 0:
                msg = " - ".join(f"{k}: {v:.4f}" for k, v in epoch_logs.items())
                print(f"Epoch {epoch + 1}/{epochs} [{time.time() - t0:.1f}s] {msg}", flush=True)
            if self.stop_training:
                break

        for cb in callbacks:
            cb.on_train_end()
        tracer.save()
        if run is not None and api.rank() == 0:
            log_model(self.module, "model")
        return history

    # ------------------------------------------------------------------ #
    def evaluate(
        self,
        data: Iterable,
        steps: Optional[int] = None,
        return_dict: bool = False,
    ):
        agg: Dict[str, float] = {}
        n = 0
        it = iter(data)
        step = 0
        while steps is None or step < steps:
            try:
                images, labels = next(it)
            except StopIteration:
                break
            logs = self.eval_step(images, labels)
            for k, v in logs.items():
                agg[k] = agg.get(k, 0.0) + v
            n += 1
            step += 1
        out = {k: v / max(n, 1) for k, v in agg.items()}
        if return_dict:
            return out
        return [out.get("loss", 0.0)] + [out[m] for m in self.metrics if m in out]

    @torch.no_grad()
    def predict(self, images: torch.Tensor, batch_size: int = 128) -> torch.Tensor:
        self.module.eval()
        outs = []
        for i in range(0, len(images), batch_size):
            outs.append(self.module(self._move(images[i : i + batch_size])).cpu())
        return torch.cat(outs)
