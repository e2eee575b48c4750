"""Single-process federated simulation + centralized trainer.

`SequentialFL` is the reference's exact execution model — `train_clients`
loops clients sequentially in one process (FLPyfhelin.py:179-198), then
encrypt -> aggregate -> decrypt -> evaluate (notebook cell 3) — minus its
shared-model bug (each client here starts from the global weights; the
reference's client i+1 silently continued from client i's weights because the
:194 reload is commented out — SURVEY.md section 7 quirks catalog).

The multi-process path (1 GPU = 1 client over RCCL) is hefl/fl/round.py;
this module is the 1-device simulation, the CLI driver and the e2e
accuracy-parity harness.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from ..config import RunConfig
from ..data.synthetic import SyntheticMedicalImages
from ..he.ckks import CKKSContext, CtxtTensor
from ..models import build_model
from .callbacks import EarlyStopping, ModelCheckpoint, ReduceLROnPlateau
from .client import LocalClient
from .metrics import classification_metrics
from .weights import flat_params, load_flat_params

# Seed offset separating held-out test data from every training stream.
TEST_SEED_OFFSET = 77777


@dataclass
class SeqRoundReport:
    metrics: Dict[str, float]
    round_seconds: float
    client_stats: List[dict] = field(default_factory=list)
    phase_seconds: Dict[str, float] = field(default_factory=dict)


class SequentialFL:
    def __init__(self, cfg: RunConfig, device: str = "cpu",
                 verbose: bool = False):
        self.cfg = cfg
        self.device = torch.device(device)
        self.verbose = verbose
        self.global_model = build_model(cfg.model, seed=cfg.fl.seed).to(self.device)
        self.clients = [LocalClient(cfg, i, device=device)
                        for i in range(cfg.fl.n_clients)]
        self.ctx: Optional[CKKSContext] = None
        self.keys = None
        if cfg.fl.encrypted:
            self.ctx = CKKSContext(cfg.he, device=device)
            self.keys = self.ctx.keygen()
        # held-out test set (reference: 400 test images, notebook cell 0/3):
        # SAME class templates as training (template_seed) so it is the same
        # task, but a disjoint seed for labels/noise — with the raw fl.seed
        # the test labels replicated the first training labels and test
        # noise could coincide with training noise, inflating the metrics.
        self.test_ds = SyntheticMedicalImages(
            cfg.fl.test_samples, cfg.model.in_shape, cfg.model.n_classes,
            seed=cfg.fl.seed + TEST_SEED_OFFSET, device=device,
            template_seed=cfg.fl.seed)
        n_train = cfg.fl.n_clients * cfg.fl.samples_per_client
        self.test_idx = torch.arange(n_train, n_train + cfg.fl.test_samples) \
            % cfg.fl.test_samples

    def _default_callbacks(self, client: LocalClient):
        # monitor validation loss when the client holds a val split (the
        # reference passes validation_data to every fit, FLPyfhelin.py:193);
        # fall back to train loss when no val samples are configured
        mon = "val_loss" if client.val_loader is not None else "loss"
        return [
            EarlyStopping(client.model, monitor=mon, patience=5,
                          restore_best=True),
            ReduceLROnPlateau(client.opt, monitor=mon, factor=0.3,
                              patience=2),
        ]

    def run_round(self, epochs: Optional[int] = None,
                  use_callbacks: bool = False) -> SeqRoundReport:
        t0 = time.perf_counter()
        g = flat_params(self.global_model)
        stats = []
        vecs = []
        for c in self.clients:
            c.set_weights(g)  # independent start from global weights
            cbs = self._default_callbacks(c) if use_callbacks else None
            s = c.local_train(epochs, callbacks=cbs)
            stats.append({"loss": s.train_loss, "accuracy": s.train_acc,
                          "steps": s.steps, "seconds": s.seconds})
            vecs.append(c.get_weights())
        t1 = time.perf_counter()
        n = len(self.clients)
        if self.ctx is not None:
            # encrypted aggregation, key-separated: sum + 1/n under pk only.
            # Same lazy-int64 bound as the distributed path (secure.py):
            # limb primes < 2**60, so > 8 summands could overflow int64.
            assert n <= 8, "lazy int64 ciphertext sum is proven for <= 8 clients"
            agg: Optional[CtxtTensor] = None
            for i, v in enumerate(vecs):
                self.ctx.reseed(i)
                ct = self.ctx.encrypt_tensor(v, self.keys.pk)
                agg = ct if agg is None else CtxtTensor(
                    agg.data + ct.data, agg.scale, agg.count)
            self.ctx.modreduce_tensor_(agg)
            avg = self.ctx.rescale_tensor(self.ctx.mul_scalar_tensor(agg, 1.0 / n))
            new_g = self.ctx.decrypt_tensor(avg, self.keys.sk).to(self.device)
        else:
            new_g = torch.stack(vecs).mean(0)
        load_flat_params(self.global_model, new_g)
        t2 = time.perf_counter()
        metrics = self.evaluate()
        t3 = time.perf_counter()
        return SeqRoundReport(
            metrics=metrics, round_seconds=t3 - t0, client_stats=stats,
            phase_seconds={"local_train": t1 - t0, "fedavg": t2 - t1,
                           "evaluate": t3 - t2})

    @torch.no_grad()
    def evaluate(self, batch_size: int = 64) -> Dict[str, float]:
        dtype = (torch.bfloat16 if self.device.type == "cuda"
                 and self.cfg.train.dtype == "bf16" else torch.float32)
        preds, trues = [], []
        idx = torch.arange(self.test_ds.n_samples)
        for i in range(0, idx.numel(), batch_size):
            x, y = self.test_ds.batch(idx[i:i + batch_size])
            logits = self.global_model(x.to(dtype))
            preds.append(logits.float().argmax(-1).cpu())
            trues.append(y.cpu())
        return classification_metrics(torch.cat(trues), torch.cat(preds),
                                      self.cfg.model.n_classes)


def train_server(cfg: RunConfig, device: str = "cpu",
                 epochs: Optional[int] = None, checkpoint_path: str = None):
    """Centralized (non-federated) trainer — the reference's train_server
    (FLPyfhelin.py:161-177, dead code there; functional here): fit on the
    full dataset with EarlyStopping / ReduceLROnPlateau / ModelCheckpoint,
    restore best, return (model, stats)."""
    import dataclasses
    solo = dataclasses.replace(cfg, fl=dataclasses.replace(cfg.fl, n_clients=1))
    client = LocalClient(solo, client_id=0, device=device)
    cbs = [
        EarlyStopping(client.model, monitor="loss", patience=3,
                      restore_best=True),
        ReduceLROnPlateau(client.opt, monitor="loss", factor=0.3, patience=2),
    ]
    if checkpoint_path:
        cbs.append(ModelCheckpoint(client.model, checkpoint_path,
                                   monitor="accuracy", mode="max"))
    stats = client.local_train(epochs, callbacks=cbs)
    return client.model, stats
