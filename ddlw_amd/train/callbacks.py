"""Training callbacks — the Keras/Horovod callback set the reference uses.

Reference call sites (SURVEY.md §2.5 "TF/Keras runtime" row):

- ``ModelCheckpoint(path, save_weights_only=True)``
  (``Part 2 .../02_hyperopt_distributed_model.py:206-211``);
- ``EarlyStopping(monitor='val_loss', min_delta=1e-2, patience=3)``
  (``Part 2 .../03_pyfunc_distributed_inference.py:397-400``);
- ``ReduceLROnPlateau(patience=10)`` (``Part 1 .../03_...py:321``);
- ``hvd.callbacks.BroadcastGlobalVariablesCallback(0)`` (:308);
- ``hvd.callbacks.MetricAverageCallback()`` (:313) — must run *before*
  LR-schedule callbacks (:310-312);
- ``hvd.callbacks.LearningRateWarmupCallback(5 epochs)`` (:315-318, Goyal et
  al. arXiv:1706.02677 linear warmup).
"""
from __future__ import annotations

import math
from pathlib import Path
from typing import Optional

import torch


class Callback:
    model = None  # set by Model.fit

    def set_model(self, model) -> None:
        self.model = model

    def on_train_begin(self, logs: Optional[dict] = None) -> None: ...

    def on_train_end(self, logs: Optional[dict] = None) -> None: ...

    def on_epoch_begin(self, epoch: int, logs: Optional[dict] = None) -> None: ...

    def on_epoch_end(self, epoch: int, logs: Optional[dict] = None) -> None: ...

    def on_batch_begin(self, batch: int, logs: Optional[dict] = None) -> None: ...

    def on_batch_end(self, batch: int, logs: Optional[dict] = None) -> None: ...


class ModelCheckpoint(Callback):
    """Rank-0-only epoch checkpoints, ``checkpoint-{epoch}.ckpt`` naming
    (reference layout)."""

    def __init__(self, filepath: str, save_weights_only: bool = True, monitor: str = "val_loss", save_best_only: bool = False):
        self.filepath = str(filepath)
        self.save_weights_only = save_weights_only
        self.monitor = monitor
        self.save_best_only = save_best_only
        self._best = math.inf

    def on_epoch_end(self, epoch: int, logs: Optional[dict] = None) -> None:
        from ..parallel import api

        if api.rank() != 0:
            return
        logs = logs or {}
        if self.save_best_only:
            cur = logs.get(self.monitor)
            if cur is None or cur >= self._best:
                return
            self._best = cur
        path = Path(self.filepath.format(epoch=epoch + 1))
        path.parent.mkdir(parents=True, exist_ok=True)
        module = self.model.module
        sd = {k: v.detach().cpu() for k, v in module.state_dict().items()}
        payload = sd if self.save_weights_only else {"state_dict": sd, "epoch": epoch}
        torch.save(payload, path)


class EarlyStopping(Callback):
    def __init__(self, monitor: str = "val_loss", min_delta: float = 0.0, patience: int = 0, mode: str = "min"):
        self.monitor = monitor
        self.min_delta = abs(min_delta)
        self.patience = patience
        self.mode = mode
        self._best = math.inf if mode == "min" else -math.inf
        self._wait = 0

    def on_epoch_end(self, epoch: int, logs: Optional[dict] = None) -> None:
        cur = (logs or {}).get(self.monitor)
        if cur is None:
            return
        improved = (cur < self._best - self.min_delta) if self.mode == "min" else (cur > self._best + self.min_delta)
        if improved:
            self._best = cur
            self._wait = 0
        else:
            self._wait += 1
            if self._wait > self.patience:
                self.model.stop_training = True


class ReduceLROnPlateau(Callback):
    def __init__(self, monitor: str = "val_loss", factor: float = 0.1, patience: int = 10, min_lr: float = 0.0):
        self.monitor = monitor
        self.factor = factor
        self.patience = patience
        self.min_lr = min_lr
        self._best = math.inf
        self._wait = 0

    def on_epoch_end(self, epoch: int, logs: Optional[dict] = None) -> None:
        cur = (logs or {}).get(self.monitor)
        if cur is None:
            return
        if cur < self._best:
            self._best = cur
            self._wait = 0
            return
        self._wait += 1
        if self._wait > self.patience:
            self._wait = 0
            for g in self.model.optimizer.param_groups:
                g["lr"] = max(g["lr"] * self.factor, self.min_lr)


class LearningRateWarmupCallback(Callback):
    """Linear per-batch warmup from ``initial_lr/size`` to ``initial_lr`` over
    ``warmup_epochs`` (Horovod semantics; Goyal et al. 1706.02677 — reference
    ``Part 1 .../03_model_training_distributed.py:315-318``)."""

    def __init__(self, warmup_epochs: int = 5, initial_lr: Optional[float] = None, verbose: bool = False):
        self.warmup_epochs = warmup_epochs
        self.initial_lr = initial_lr
        self._steps_per_epoch = None
        self._batch = 0

    def on_train_begin(self, logs: Optional[dict] = None) -> None:
        from ..parallel import api

        if self.initial_lr is None:
            self.initial_lr = self.model.optimizer.param_groups[0]["lr"]
        self._size = max(api.size(), 1)
        self._steps_per_epoch = self.model._steps_per_epoch or 1
        self._batch = 0

    def on_batch_begin(self, batch: int, logs: Optional[dict] = None) -> None:
        total = self.warmup_epochs * self._steps_per_epoch
        if self._batch >= total or total == 0:
            return
        start = self.initial_lr / self._size
        frac = (self._batch + 1) / total
        lr = start + (self.initial_lr - start) * frac
        for g in self.model.optimizer.param_groups:
            g["lr"] = lr
        self._batch += 1


class BroadcastGlobalVariablesCallback(Callback):
    """Broadcast model + optimizer state from root at train start (C3)."""

    def __init__(self, root_rank: int = 0):
        self.root_rank = root_rank

    def on_train_begin(self, logs: Optional[dict] = None) -> None:
        from ..parallel import api

        api.broadcast_parameters(self.model.module, self.root_rank)
        api.broadcast_optimizer_state(self.model.optimizer, self.root_rank)


class MetricAverageCallback(Callback):
    """Average epoch metrics across ranks before schedule callbacks read them
    (C4). Model.fit orders this before LR callbacks, as the reference notes
    (``.../03_model_training_distributed.py:310-313``)."""

    def on_epoch_end(self, epoch: int, logs: Optional[dict] = None) -> None:
        from ..parallel import api

        if logs:
            logs.update(api.allreduce_metrics(logs))
