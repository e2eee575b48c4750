"""Per-client local training.

Replaces the reference's `train_clients` (FLPyfhelin.py:179-198). Deliberate
behavioral delta (SURVEY.md section 7 quirks catalog): each client trains an
INDEPENDENT model initialized from the global weights each round — the
reference's shared-model-across-clients bug (FLPyfhelin.py:180, the :194
reload is commented out) is not reproduced.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, Optional

import torch

from ..config import RunConfig
from ..data.synthetic import ClientLoader, SyntheticMedicalImages
from ..data.shard import shard_indices
from ..models import build_model
from ..ops.adam import FusedAdam
from ..ops.functional import softmax_xent
from .weights import flat_params, load_flat_params


@dataclass
class RoundStats:
    train_loss: float = 0.0
    train_acc: float = 0.0
    samples: int = 0
    steps: int = 0
    seconds: float = 0.0
    phase_seconds: Dict[str, float] = field(default_factory=dict)


class LocalClient:
    """One FL client: a model + optimizer + data shard on one device."""

    def __init__(self, cfg: RunConfig, client_id: int, device: str = "cpu"):
        self.cfg = cfg
        self.client_id = client_id
        self.device = torch.device(device)
        self.model = build_model(cfg.model, seed=cfg.fl.seed).to(self.device)
        t = cfg.train
        self.opt = FusedAdam(self.model.parameters(), lr=t.lr, decay=t.lr_decay,
                             beta1=t.beta1, beta2=t.beta2, eps=t.eps)
        self.compute_dtype = (torch.bfloat16 if (self.device.type == "cuda"
                              and t.dtype == "bf16") else torch.float32)
        self.use_graphs = (self.device.type == "cuda"
                           and getattr(t, "hip_graphs", True))
        self._graphs = {}
        # Data: the full federated dataset is conceptually
        # n_clients * (train + val) samples; this client materializes its
        # contiguous shard (reference sharding semantics, FLPyfhelin.py:75-78)
        # and splits off the TRAILING val fraction (reference:
        # validation_split=0.1 of an 800-sample shard -> 720 train / 80 val,
        # FLPyfhelin.py:85-99). Training runs on the train subset only;
        # val metrics feed the callbacks (fit(validation_data=...), :193).
        per_client = cfg.fl.samples_per_client + cfg.fl.val_samples_per_client
        n_total = cfg.fl.n_clients * per_client
        self.dataset = SyntheticMedicalImages(
            n_total, cfg.model.in_shape, cfg.model.n_classes,
            seed=cfg.fl.seed, device=device, dtype=self.compute_dtype)
        idx = shard_indices(n_total, client_id, cfg.fl.n_clients)
        n_val = cfg.fl.val_samples_per_client
        train_idx = idx[: idx.numel() - n_val]
        # augmentation applies to TRAINING batches only (the reference's
        # ImageDataGenerator transforms train, not val/test)
        affine = {"none": None, "hflip": (0.0, 0.0, True),
                  "full": (0.2, 0.2, True)}[cfg.fl.augment]
        self.affine = affine
        self.loader = ClientLoader(self.dataset, train_idx, t.batch_size,
                                   seed=cfg.fl.seed + client_id,
                                   affine=affine)
        self.val_loader = (ClientLoader(self.dataset,
                                        idx[idx.numel() - n_val:],
                                        t.batch_size, shuffle=False)
                           if n_val else None)

    def train_step(self, x: torch.Tensor, y: torch.Tensor):
        if self.use_graphs:
            return self._graphed_step(x, y)
        return self._eager_step(x, y)

    def _eager_step(self, x: torch.Tensor, y: torch.Tensor):
        x = x.to(self.compute_dtype)
        logits = self.model(x)
        loss = softmax_xent(logits, y)
        self.opt.zero_grad_() if self.use_graphs else self.opt.zero_grad()
        loss.backward()
        self.opt.step()
        return loss, logits

    # ----- hipGraph-captured training step (MI355X): one graph replay per
    # step instead of ~40 eager kernel launches + Python autograd overhead.
    # One graph per batch shape (the last shard batch may be partial). -----
    def _graphed_step(self, x: torch.Tensor, y: torch.Tensor):
        key = tuple(x.shape)
        ent = self._graphs.get(key)
        if ent is None:
            ent = self._capture(x, y)
            self._graphs[key] = ent
        ent["x"].copy_(x)
        ent["y"].copy_(y)
        ent["graph"].replay()
        # Adam as two direct launches against this graph's grad pointers —
        # the captured backward STEALS fresh pooled grad tensors (stable
        # addresses across replays), so no zero/accumulate kernels exist
        self.opt.step_mt(ent["mt"])
        return ent["loss"], ent["logits"]

    def _capture(self, x: torch.Tensor, y: torch.Tensor):
        sx, sy = x.clone(), y.clone()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):  # warmup (real steps; grads stay materialized)
                self._eager_warmup(sx, sy)
        torch.cuda.current_stream().wait_stream(side)
        # device-side Adam schedule buffers, created OUTSIDE the graph and
        # seeded with the warmup step count
        self.opt.prepare_graph_state(sx.device)
        g = torch.cuda.CUDAGraph()
        # capture on the SAME stream the warmup ran on: AccumulateGrad nodes
        # bind to the stream that first materialized each .grad, and a capture
        # on a different stream lets the accumulation escape the graph
        # (symptom: replays run but weights never learn)
        if not hasattr(self, "_acc_loss"):
            self._acc_loss = torch.zeros((), dtype=torch.float32, device=sx.device)
            self._acc_correct = torch.zeros((), dtype=torch.float32,
                                            device=sx.device)
            self._one = torch.ones((), dtype=torch.float32, device=sx.device)
        # bf16 shadow weights must exist BEFORE capture so the captured
        # forward reads them (instead of recording per-step cast kernels)
        self.opt.ensure_shadows()
        # grads set to None: the captured backward steals fresh grad tensors
        # from the graph pool; replays rewrite them in place
        self.opt.zero_grad()
        with torch.cuda.graph(g, stream=side):
            logits = self.model(sx.to(self.compute_dtype))
            # stats accumulate INSIDE the loss kernel into persistent buffers
            loss = softmax_xent(logits, sy, self._acc_loss, self._acc_correct)
            # persistent gradient seed: no per-step ones() fill kernel
            loss.backward(gradient=self._one)
        mt = self.opt.build_mt_table()  # this graph's stolen-grad pointers
        # hold the stolen grad tensors: a later capture re-steals p.grad and
        # would otherwise drop the refs, returning these pool blocks to the
        # allocator while this graph's kernels still write them
        grads = [p.grad for p in self.opt.params]
        return {"graph": g, "x": sx, "y": sy, "loss": loss, "logits": logits,
                "mt": mt, "grads": grads}

    def _eager_warmup(self, x, y, clear_grads=False):
        self.opt.zero_grad()  # grads=None: fresh tensors each warmup step
        logits = self.model(x.to(self.compute_dtype))
        loss = softmax_xent(logits, y)
        loss.backward()
        self.opt.step()
        if clear_grads:
            # fill-skip contract: the shared grad_buf accumulation buffers
            # must leave every step zeroed (in-graph the Adam zero_g pass
            # does this; eager warmup does it explicitly)
            with torch.no_grad():
                for p in self.opt.params:
                    if p.grad is not None:
                        p.grad.zero_()

    # ----- whole-epoch hipGraph: ALL steps of one local epoch (forward,
    # loss, backward, Adam) captured as ONE graph. Per epoch the host does
    # one data-stage + one replay instead of ~4 Python dispatches per step —
    # on CNN2-sized models the per-step Python/launch overhead dominates,
    # so this is the main lever on the headline FL-rounds/sec metric.
    # RNG stays OUTSIDE the graph: the epoch's samples are generated and
    # staged before each replay; the graph reads static slices. One grad
    # pointer table serves every captured step because the Adam kernel
    # zeroes each grad as it consumes it (consume-and-clear), so the next
    # step's backward accumulates into zeroed, address-stable buffers. -----
    def _ensure_epoch_graph(self):
        ent = getattr(self, "_ep_ent", None)
        if ent is not None:
            return ent
        B = self.loader.batch_size
        n = self.loader.indices.numel()
        # in-graph data staging: the captured graph gathers labels and
        # generates the epoch's samples itself (synth kernel reads its seed
        # from a device buffer the host rewrites before each replay), so
        # per epoch the host does ONE small H2D copy + ONE replay. The
        # affine augmentation is fused INTO the synth kernel, so it stays
        # in-graph; only a custom augment hook or non-bf16 dtype falls back
        # to host staging (those paths go through the torch ops).
        in_graph_data = (self.loader.augment is None
                         and self.compute_dtype == torch.bfloat16)
        x0, y0 = self.dataset.batch(self.loader.indices)
        if self.loader.augment is not None:
            x0 = self.loader.augment(x0)
        X = x0.to(self.compute_dtype).contiguous()
        Y = y0.contiguous()
        from ..ops import functional as Fx
        import os
        no_zero = os.getenv("HEFL_GRAPH_NO_ZERO", "1") == "1"
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            try:
                # warmup runs under the fill-skip flag too: the grad_buf
                # allocations (cnn.hip) must happen HERE, eagerly — during
                # capture hipMalloc is illegal and grad_buf would fall
                # back to the classic zeroed path for the whole graph
                Fx.GRAPH_NO_ZERO = no_zero
                for _ in range(3):  # warmup on the full-batch shape
                    self._eager_warmup(X[:B], Y[:B], clear_grads=no_zero)
                if n % B:  # and once on the partial tail-batch shape
                    self._eager_warmup(X[n - (n % B):], Y[n - (n % B):],
                                       clear_grads=no_zero)
            finally:
                Fx.GRAPH_NO_ZERO = False
        torch.cuda.current_stream().wait_stream(side)
        self.opt.prepare_graph_state(X.device)
        if not hasattr(self, "_acc_loss"):
            self._acc_loss = torch.zeros((), dtype=torch.float32, device=X.device)
            self._acc_correct = torch.zeros((), dtype=torch.float32,
                                            device=X.device)
            self._one = torch.ones((), dtype=torch.float32, device=X.device)
        self.opt.ensure_shadows()
        # Per-step pointer tables: each captured backward STEALS fresh
        # pooled grad tensors (no zero/accumulate kernels at all); the
        # tables the captured Adam kernels read are filled AFTER capture
        # with the addresses recorded during it (capture-pool addresses
        # are replay-stable). One schedule-prep kernel serves the epoch.
        steps = (n + B - 1) // B
        shells = [self.opt.alloc_mt_shell() for _ in range(steps)]
        rows_per_step = []
        grads_hold = []
        order_buf = torch.empty(n, dtype=torch.int64, device=self.device)
        seed_buf = torch.zeros(1, dtype=torch.int64, device=self.device)
        order_buf.copy_(self.loader.indices.to(self.device))
        g = torch.cuda.CUDAGraph()
        import hefl
        C = hefl.load_extension()
        # HEFL_GRAPH_NO_ZERO (default ON): stolen grads skip their
        # zero-init fills — the captured Adam clears each grad as it
        # consumes it (zero_grad=True below) and the eager zero_() after
        # capture covers the first replay. SOUND because the accumulation
        # buffers are process-lifetime hipMalloc allocations keyed by
        # param id (cnn.hip grad_buf), never part of the caching
        # allocator: nothing allocated during capture can alias them. The
        # first design accumulated into capture-pool tensors instead and
        # diverged on resnet18 when a grad reused a block freed earlier
        # in the same capture — that hazard is what grad_buf removes.
        # Measured (sound build): config2 36.7->38.4 rounds/s, reference
        # 4.8->5.03 (the fills were 11.5% of config2 kernel time).
        try:
            Fx.GRAPH_NO_ZERO = no_zero
            with torch.cuda.graph(g, stream=side):
                if in_graph_data:
                    Y = self.dataset.labels.index_select(0, order_buf)
                    zr, sr, fl = self.affine or (0.0, 0.0, False)
                    X = C.synth_batch_g(self.dataset.templates, Y, seed_buf, 0,
                                        zoom=zr, shear=sr, flip=int(bool(fl)))
                self.opt.prep_epoch(steps)
                for s, i in enumerate(range(0, n, B)):
                    self.opt.zero_grad()  # grads=None -> backward steals
                    logits = self.model(X[i:i + B])
                    loss = softmax_xent(logits, Y[i:i + B], self._acc_loss,
                                        self._acc_correct)
                    loss.backward(gradient=self._one)
                    rows_per_step.append(self.opt.current_ptr_rows())
                    self.opt.step_mt_at(shells[s], s, zero_grad=no_zero)
                    grads_hold.append([p.grad for p in self.opt.params])
        finally:
            Fx.GRAPH_NO_ZERO = False
        with torch.no_grad():
            for step_grads in grads_hold:
                for t in step_grads:
                    t.zero_()
        for shell, rows in zip(shells, rows_per_step):
            self.opt.fill_mt_shell(shell, rows)
        ent = {"graph": g, "X": X, "Y": Y, "mt": shells,
               "grads": grads_hold, "steps": steps, "n": n,
               "order_buf": order_buf, "seed_buf": seed_buf,
               "in_graph_data": in_graph_data, "seed_ctr": 0}
        self._ep_ent = ent
        return ent

    def _epoch_replay(self, ent) -> None:
        """Stage one (shuffled) epoch of data, then replay the epoch graph."""
        order = self.loader.epoch_order()
        if ent["in_graph_data"]:
            # data generation is captured: host work is two small copies
            ent["order_buf"].copy_(order.to(self.device))
            ent["seed_ctr"] += 1
            ent["seed_buf"].fill_(self.dataset.seed * 0x20003
                                  + ent["seed_ctr"])
            ent["graph"].replay()
            return
        x, y = self.dataset.batch(order, affine=self.loader.affine)
        if self.loader.augment is not None:
            x = self.loader.augment(x)
        ent["X"].copy_(x.to(self.compute_dtype))
        ent["Y"].copy_(y)
        ent["graph"].replay()

    def local_train(self, epochs: Optional[int] = None,
                    callbacks: Optional[list] = None) -> RoundStats:
        """Local SGD for `epochs` epochs. With callbacks (the reference wires
        EarlyStopping/ReduceLROnPlateau/ModelCheckpoint into every fit,
        FLPyfhelin.py:186-192) per-epoch metrics sync to host; without,
        the loop is fully sync-free until the end."""
        epochs = self.cfg.train.local_epochs if epochs is None else epochs
        callbacks = callbacks or []
        stats = RoundStats()
        dev = self.device
        graphed_stats = self.use_graphs and not callbacks
        if graphed_stats and hasattr(self, "_acc_loss"):
            self._acc_loss.zero_()
            self._acc_correct.zero_()
        loss_sum = torch.zeros((), dtype=torch.float32, device=dev)
        acc_sum = torch.zeros((), dtype=torch.float32, device=dev)
        t0 = time.perf_counter()
        ep_ent = self._ensure_epoch_graph() if graphed_stats else None
        for ep in range(epochs):
            if graphed_stats:
                # whole epoch = one staged-data copy + one graph replay;
                # stats accumulate inside the captured loss kernels
                self._epoch_replay(ep_ent)
                stats.steps += ep_ent["steps"]
                stats.samples += ep_ent["n"]
                continue
            ep_loss = torch.zeros((), dtype=torch.float32, device=dev)
            ep_acc = torch.zeros((), dtype=torch.float32, device=dev)
            ep_steps = ep_samples = 0
            for x, y in self.loader:
                loss, logits = self.train_step(x, y)
                ep_steps += 1
                ep_samples += y.numel()
                # device-side accumulation: no per-step host sync
                ep_loss += loss.detach().float()
                ep_acc += (logits.detach().float().argmax(-1) == y).float().sum()
            loss_sum += ep_loss
            acc_sum += ep_acc
            stats.steps += ep_steps
            stats.samples += ep_samples
            if callbacks:
                logs = {"loss": float(ep_loss) / max(ep_steps, 1),
                        "accuracy": float(ep_acc) / max(ep_samples, 1)}
                if self.val_loader is not None:
                    vl, va = self.evaluate_loss(self.val_loader)
                    logs["val_loss"], logs["val_accuracy"] = vl, va
                for cb in callbacks:
                    cb.on_epoch_end(ep, logs)
                if any(cb.stop_training for cb in callbacks):
                    break
        for cb in callbacks:
            cb.on_train_end()
        if dev.type == "cuda":
            torch.cuda.synchronize()
        stats.seconds = time.perf_counter() - t0
        if stats.steps:
            if graphed_stats and hasattr(self, "_acc_loss"):
                stats.train_loss = float(self._acc_loss) / stats.steps
                stats.train_acc = float(self._acc_correct) / max(stats.samples, 1)
            else:
                stats.train_loss = float(loss_sum) / stats.steps
                stats.train_acc = float(acc_sum) / max(stats.samples, 1)
        return stats

    # ----- FedAvg weight transport: on GPU, ONE multi-tensor kernel packs
    # (or unpacks + refreshes bf16 shadows) the whole parameter vector,
    # replacing ~3 kernels per parameter tensor. -----
    def _weight_tables(self):
        wt = getattr(self, "_wt", None)
        if wt is not None:
            return wt
        import hefl  # extension must be present on GPU (fails loudly if not)
        hefl.load_extension()
        from .weights import _agg_tensors
        self.opt.ensure_shadows()
        dev = self.device
        tensors = list(_agg_tensors(self.model))
        ptrs, shp, sizes, offs, meta = [], [], [], [], []
        total = 0
        for t, p in enumerate(tensors):
            assert p.dtype == torch.float32 and p.is_contiguous()
            ptrs.append(p.data_ptr())
            sh = getattr(p, "_bf16", None)
            shp.append(sh.data_ptr() if sh is not None else 0)
            sizes.append(p.numel())
            offs.append(total)
            total += p.numel()
            for off in range(0, p.numel(), FusedAdam._MT_CHUNK):
                meta.append([t, off])
        self._wt = {
            "meta": torch.tensor(meta, dtype=torch.int64, device=dev),
            "ptrs": torch.tensor(ptrs, dtype=torch.int64, device=dev),
            "shptrs": torch.tensor(shp, dtype=torch.int64, device=dev),
            "sizes": torch.tensor(sizes, dtype=torch.int64, device=dev),
            "offs": torch.tensor(offs, dtype=torch.int64, device=dev),
            "n": len(meta), "total": total,
        }
        return self._wt

    def get_weights(self) -> torch.Tensor:
        if self.device.type == "cuda":
            import hefl
            wt = self._weight_tables()
            flat = torch.empty(wt["total"], dtype=torch.float32,
                               device=self.device)
            hefl.load_extension().pack_mt(wt["meta"], wt["ptrs"], wt["sizes"],
                                          wt["offs"], wt["n"], flat)
            return flat
        return flat_params(self.model)

    def set_weights(self, vec: torch.Tensor) -> None:
        if self.device.type == "cuda":
            import hefl
            wt = self._weight_tables()
            v = vec.to(device=self.device, dtype=torch.float32).contiguous()
            if v.numel() != wt["total"]:
                raise ValueError(f"vector length {v.numel()} != {wt['total']}")
            hefl.load_extension().unpack_mt(v, wt["meta"], wt["ptrs"],
                                            wt["shptrs"], wt["sizes"],
                                            wt["offs"], wt["n"])
            return
        load_flat_params(self.model, vec.to(self.device))
        if hasattr(self.opt, "refresh_shadows"):
            self.opt.refresh_shadows()  # bf16 shadows must follow FedAvg loads

    @torch.no_grad()
    def evaluate_loss(self, loader: ClientLoader):
        """Mean loss + accuracy over a loader (the per-epoch validation pass
        the reference gets from fit(validation_data=...), FLPyfhelin.py:193)."""
        dev = self.device
        loss_sum = torch.zeros((), dtype=torch.float32, device=dev)
        acc_sum = torch.zeros((), dtype=torch.float32, device=dev)
        steps = samples = 0
        for x, y in loader:
            logits = self.model(x.to(self.compute_dtype))
            loss_sum += softmax_xent(logits, y).detach().float()
            acc_sum += (logits.detach().float().argmax(-1) == y).float().sum()
            steps += 1
            samples += y.numel()
        return (float(loss_sum) / max(steps, 1),
                float(acc_sum) / max(samples, 1))

    @torch.no_grad()
    def evaluate(self, dataset: SyntheticMedicalImages, indices: torch.Tensor,
                 batch_size: int = 64):
        """Return (y_true, y_pred) over the given samples."""
        preds, trues = [], []
        for i in range(0, indices.numel(), batch_size):
            x, y = dataset.batch(indices[i:i + batch_size])
            logits = self.model(x.to(self.compute_dtype))
            preds.append(logits.float().argmax(-1).cpu())
            trues.append(y.cpu())
        return torch.cat(trues), torch.cat(preds)
