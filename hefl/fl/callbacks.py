"""Training callbacks: the reference's Keras callback set re-implemented.

The reference wires EarlyStopping(monitor='loss', restore_best_weights),
ReduceLROnPlateau(monitor='loss', factor, patience) and ModelCheckpoint
(save_best_only, monitor accuracy) into every fit call
(FLPyfhelin.py:163-171, 186-192). Same semantics here, driven per-epoch by
LocalClient.local_train / train_server.
"""
from __future__ import annotations

import copy
from typing import Dict

import torch


class Callback:
    def on_epoch_end(self, epoch: int, logs: Dict[str, float]) -> None: ...
    @property
    def stop_training(self) -> bool:
        return False

    def on_train_end(self) -> None: ...


class EarlyStopping(Callback):
    """Stop when `monitor` has not improved for `patience` epochs; optionally
    restore the best weights (reference: patience 3 or 5, restore_best=True,
    FLPyfhelin.py:163-165,186-188 — note the reference never actually reloads
    best weights after fit because :194 is commented out; restoring here is a
    deliberate fix, SURVEY.md section 7 quirks catalog)."""

    def __init__(self, model: torch.nn.Module, monitor: str = "loss",
                 patience: int = 5, restore_best: bool = True,
                 mode: str = "min"):
        self.model = model
        self.monitor = monitor
        self.patience = patience
        self.restore_best = restore_best
        self.sign = 1.0 if mode == "min" else -1.0
        self.best = float("inf")
        self.best_state = None
        self.wait = 0
        self._stop = False

    def on_epoch_end(self, epoch, logs):
        cur = self.sign * logs[self.monitor]
        if cur < self.best:
            self.best = cur
            self.wait = 0
            if self.restore_best:
                self.best_state = copy.deepcopy(
                    {k: v.detach().clone() for k, v in self.model.state_dict().items()})
        else:
            self.wait += 1
            if self.wait >= self.patience:
                self._stop = True

    @property
    def stop_training(self):
        return self._stop

    def on_train_end(self):
        if self.restore_best and self.best_state is not None:
            self.model.load_state_dict(self.best_state)


class ReduceLROnPlateau(Callback):
    """Multiply the optimizer lr by `factor` after `patience` stale epochs
    (reference: factor 0.3, patience 2, FLPyfhelin.py:166,189)."""

    def __init__(self, optimizer, monitor: str = "loss", factor: float = 0.3,
                 patience: int = 2, min_lr: float = 1e-6, mode: str = "min"):
        self.opt = optimizer
        self.monitor = monitor
        self.factor = factor
        self.patience = patience
        self.min_lr = min_lr
        self.sign = 1.0 if mode == "min" else -1.0
        self.best = float("inf")
        self.wait = 0

    def on_epoch_end(self, epoch, logs):
        cur = self.sign * logs[self.monitor]
        if cur < self.best:
            self.best = cur
            self.wait = 0
        else:
            self.wait += 1
            if self.wait >= self.patience:
                self.opt.set_lr(max(self.opt.lr * self.factor, self.min_lr))
                self.wait = 0


class ModelCheckpoint(Callback):
    """Save weights when `monitor` improves (reference: best accuracy,
    save_weights_only, FLPyfhelin.py:167-171,190-192)."""

    def __init__(self, model: torch.nn.Module, path: str,
                 monitor: str = "accuracy", mode: str = "max"):
        self.model = model
        self.path = path
        self.monitor = monitor
        self.sign = -1.0 if mode == "max" else 1.0
        self.best = float("inf")

    def on_epoch_end(self, epoch, logs):
        cur = self.sign * logs[self.monitor]
        if cur < self.best:
            self.best = cur
            from .checkpoint import save_model_weights
            save_model_weights(self.model, self.path)
