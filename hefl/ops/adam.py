"""Adam with Keras-style lr decay (lr_t = lr / (1 + decay * t)).

Matches the reference's optimizer: Adam(learning_rate=INIT_LR,
decay=INIT_LR/10) with epsilon=1e-7 (FLPyfhelin.py:140; Keras defaults).
GPU path is one fused HIP kernel per parameter tensor (m, v update, bias
correction, decayed lr, parameter write) — replacing TF's Adam kernels
(Adam row).
"""
from __future__ import annotations

from typing import Iterable

import torch

import hefl


class FusedAdam:
    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float = 1e-3,
                 decay: float = 1e-4, beta1: float = 0.9, beta2: float = 0.999,
                 eps: float = 1e-7):
        self.params = [p for p in params if p.requires_grad]
        self.lr, self.decay = lr, decay
        self.beta1, self.beta2, self.eps = beta1, beta2, eps
        self.step_count = 0
        self.m = [torch.zeros_like(p, dtype=torch.float32) for p in self.params]
        self.v = [torch.zeros_like(p, dtype=torch.float32) for p in self.params]

    @torch.no_grad()
    def step(self):
        self.step_count += 1
        t = self.step_count
        lr_t = self.lr / (1.0 + self.decay * (t - 1))  # Keras decay schedule
        bc1 = 1.0 - self.beta1 ** t
        bc2 = 1.0 - self.beta2 ** t
        for p, m, v in zip(self.params, self.m, self.v):
            if p.grad is None:
                continue
            g = p.grad
            if p.is_cuda:
                hefl.load_extension().fused_adam(
                    p.data, g, m, v, lr_t, self.beta1, self.beta2, self.eps, bc1, bc2)
                sh = getattr(p, "_bf16", None)
                if sh is not None:
                    sh.copy_(p.data.to(torch.bfloat16))
            else:
                g = g.float()
                m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
                v.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
                mh = m / bc1
                vh = v / bc2
                p.data.add_(-lr_t * mh / (vh.sqrt() + self.eps))

    _MT_CHUNK = 2048  # matches MT_CHUNK in hefl/csrc/cnn.hip

    def prepare_graph_state(self, device=None):
        """Create device-side step/schedule/hyper buffers plus the
        multi-tensor chunk table (must run OUTSIDE any hipGraph capture,
        with .grad materialized). Created ONCE: captured graphs hold raw
        pointers to these buffers, so re-creating them would leave earlier
        graphs reading freed memory."""
        if hasattr(self, "_step_t"):
            return  # keep live buffers; device counter keeps running
        dev = device if device is not None else self.params[0].device
        self._step_t = torch.tensor([self.step_count], dtype=torch.int64,
                                    device=dev)
        self._sched = torch.zeros(3, dtype=torch.float32, device=dev)
        self._hyper = torch.tensor([self.lr, self.decay],
                                   dtype=torch.float32, device=dev)

    def set_lr(self, lr: float):
        self.lr = lr
        if hasattr(self, "_hyper"):
            self._hyper[0] = lr  # reaches captured graphs without re-capture

    def set_step(self, step: int):
        """Set the step counter, keeping the device-side counter captured
        graphs read in sync (checkpoint resume on a graphed client)."""
        self.step_count = int(step)
        if hasattr(self, "_step_t"):
            self._step_t.fill_(int(step))

    def build_mt_table(self):
        """Multi-tensor chunk table from the CURRENT .grad pointers — built
        per captured graph (each capture's backward steals fresh pooled grad
        tensors, so every graph gets its own table)."""
        assert all(p.grad is not None for p in self.params), \
            "grads must be materialized before build_mt_table"
        dev = self.params[0].device
        self.ensure_shadows()
        ptrs, sizes, meta = [], [], []
        for t, (p, m, v) in enumerate(zip(self.params, self.m, self.v)):
            sh = getattr(p, "_bf16", None)
            ptrs.append([p.data.data_ptr(), p.grad.data_ptr(),
                         m.data_ptr(), v.data_ptr(),
                         sh.data_ptr() if sh is not None else 0])
            sizes.append(p.numel())
            for off in range(0, p.numel(), self._MT_CHUNK):
                meta.append([t, off])
        return {"ptrs": torch.tensor(ptrs, dtype=torch.int64, device=dev),
                "sizes": torch.tensor(sizes, dtype=torch.int64, device=dev),
                "meta": torch.tensor(meta, dtype=torch.int64, device=dev),
                "n": len(meta)}

    def ensure_shadows(self):
        """bf16 shadow weights: the forward kernels read these instead of
        casting the fp32 masters every step; the multi-tensor Adam kernel
        refreshes them in the same pass that updates the masters."""
        for p in self.params:
            if p.is_cuda and getattr(p, "_bf16", None) is None:
                p._bf16 = p.detach().to(torch.bfloat16).contiguous()

    def refresh_shadows(self):
        """Re-sync shadows after an external weight write (FedAvg load)."""
        for p in self.params:
            sh = getattr(p, "_bf16", None)
            if sh is not None:
                sh.copy_(p.detach().to(torch.bfloat16))

    @torch.no_grad()
    def step_mt(self, table, zero_grad: bool = False):
        """Two direct kernel launches (schedule advance + one multi-tensor
        Adam over all params) against a per-graph pointer table. Runs AFTER
        a graph replay — the replay wrote the grads the table points at.
        With zero_grad the Adam kernel clears each grad as it consumes it."""
        C = hefl.load_extension()
        C.adam_prep(self._step_t, self._sched, self._hyper,
                    self.beta1, self.beta2)
        C.fused_adam_mt(table["meta"], table["ptrs"], table["sizes"],
                        table["n"], self._sched, self.beta1, self.beta2,
                        self.eps, 1 if zero_grad else 0, 0)

    # ----- epoch-graph mode: S captured steps share ONE schedule-prep
    # kernel (S rows of (lr, bc1, bc2)) and get per-step pointer tables
    # whose contents are filled AFTER capture (the captured backward steals
    # fresh pooled grad tensors per step; addresses are replay-stable). -----
    def prep_epoch(self, steps: int):
        """One kernel: schedule rows for `steps` captured steps; advances
        the device step counter by `steps` per replay."""
        if not hasattr(self, "_sched_ep") or self._sched_ep.numel() < 3 * steps:
            self._sched_ep = torch.zeros(3 * steps, dtype=torch.float32,
                                         device=self.params[0].device)
        hefl.load_extension().adam_prep_epoch(
            self._step_t, self._sched_ep, self._hyper, self.beta1, self.beta2,
            steps)

    def _mt_shared(self):
        if not hasattr(self, "_mt_meta"):
            dev = self.params[0].device
            sizes, meta = [], []
            for t, p in enumerate(self.params):
                sizes.append(p.numel())
                for off in range(0, p.numel(), self._MT_CHUNK):
                    meta.append([t, off])
            self._mt_meta = torch.tensor(meta, dtype=torch.int64, device=dev)
            self._mt_sizes = torch.tensor(sizes, dtype=torch.int64, device=dev)
        return self._mt_meta, self._mt_sizes

    def alloc_mt_shell(self):
        """Empty per-step pointer table; captured kernels reference its
        (stable) storage, contents are written post-capture."""
        meta, sizes = self._mt_shared()
        ptrs = torch.empty(len(self.params), 5, dtype=torch.int64,
                           device=self.params[0].device)
        return {"ptrs": ptrs, "meta": meta, "sizes": sizes,
                "n": meta.shape[0]}

    def current_ptr_rows(self):
        """Snapshot {p, grad, m, v, shadow} device addresses (called during
        capture, right after a step's backward stole its grad tensors)."""
        self.ensure_shadows()
        rows = []
        for p, m, v in zip(self.params, self.m, self.v):
            sh = getattr(p, "_bf16", None)
            rows.append([p.data.data_ptr(), p.grad.data_ptr(), m.data_ptr(),
                         v.data_ptr(), sh.data_ptr() if sh is not None else 0])
        return rows

    def fill_mt_shell(self, shell, rows):
        shell["ptrs"].copy_(torch.tensor(rows, dtype=torch.int64))

    @torch.no_grad()
    def step_mt_at(self, table, sched_off: int, zero_grad: bool = False):
        """Adam-only launch (no prep) against the epoch schedule row
        `sched_off` — captured once per step inside the epoch graph.
        zero_grad: consume-and-clear each grad (required when the backward
        skipped its zero-init fills — functional.GRAPH_NO_ZERO)."""
        C = hefl.load_extension()
        C.fused_adam_mt(table["meta"], table["ptrs"], table["sizes"],
                        table["n"], self._sched_ep, self.beta1, self.beta2,
                        self.eps, 1 if zero_grad else 0, sched_off)

    def zero_grad(self):
        for p in self.params:
            p.grad = None

    def zero_grad_(self):
        """In-place grad zeroing (keeps buffers alive)."""
        for p in self.params:
            if p.grad is not None:
                p.grad.zero_()

    def state_dict(self):
        step = self.step_count
        if hasattr(self, "_step_t"):
            step = int(self._step_t.item())  # graphed steps advance on device
        return {"step": step, "m": self.m, "v": self.v,
                "lr": self.lr, "decay": self.decay}

    def load_state_dict(self, sd):
        self.set_step(sd["step"])
        if "lr" in sd:
            self.set_lr(sd["lr"])
        if "decay" in sd:
            self.decay = sd["decay"]
            if hasattr(self, "_hyper"):
                self._hyper[1] = sd["decay"]
        for dst, src in zip(self.m, sd["m"]):
            dst.copy_(src)
        for dst, src in zip(self.v, sd["v"]):
            dst.copy_(src)
