"""Synthetic medical-image-shaped data.

The reference reads chest-image folders through Keras ImageDataGenerator
(FLPyfhelin.py:57-114). The north star uses synthetic data (no network for
datasets), so this module generates class-conditional images on the fly:
label-dependent Gaussian blobs + noise, deterministic in (seed, index), in
NHWC layout and [0, 1] range (the reference rescales 1/255, FLPyfhelin.py:59).
"""
from __future__ import annotations

from typing import Iterator, Tuple

import torch


class SyntheticMedicalImages:
    """Deterministic synthetic dataset of (image NHWC, label).

    Images are produced in batches directly on the requested device so the
    training loop never touches the host (reference equivalent: the
    ImageDataGenerator pipeline, FLPyfhelin.py:80-99).
    """

    def __init__(self, n_samples: int, in_shape: Tuple[int, int, int],
                 n_classes: int, seed: int = 0, device: str = "cpu",
                 dtype: torch.dtype = torch.float32,
                 template_seed: int = None):
        """`template_seed` pins the per-class templates independently of
        `seed`: a held-out TEST set must share the train templates (same
        classification task) while drawing disjoint labels and noise —
        pass template_seed=<train seed>, seed=<train seed + offset>."""
        self.n_samples = int(n_samples)
        self.H, self.W, self.C = in_shape
        self.n_classes = int(n_classes)
        self.seed = int(seed)
        self.device = torch.device(device)
        self.dtype = dtype
        # Per-class signal template: fixed low-frequency pattern per class so a
        # CNN can actually learn (accuracy-parity validation mirrors the
        # reference's end-to-end statistical check).
        g = torch.Generator(device="cpu").manual_seed(
            (self.seed if template_seed is None else int(template_seed))
            ^ 0x5EED)
        base = torch.randn(self.n_classes, 8, 8, self.C, generator=g)
        self.templates = torch.nn.functional.interpolate(
            base.permute(0, 3, 1, 2), size=(self.H, self.W), mode="bilinear",
            align_corners=False).permute(0, 2, 3, 1).contiguous()
        self.templates = self.templates.to(self.device)
        if template_seed is not None:
            g = torch.Generator(device="cpu").manual_seed(self.seed ^ 0x1AB315)
        self.labels = torch.randint(0, self.n_classes, (self.n_samples,),
                                    generator=g).to(self.device)

    def batch(self, indices: torch.Tensor,
              affine=None) -> Tuple[torch.Tensor, torch.Tensor]:
        """Return (x[NHWC], y[N]) for the given sample indices.

        `affine` = (zoom_range, shear_range, hflip) applies a per-sample
        random zoom/shear/horizontal-flip when sampling the class template
        — the analog of the reference's ImageDataGenerator augmentation
        (shear 0.2, zoom 0.2, horizontal_flip; FLPyfhelin.py:80-86)."""
        idx = indices.to(self.device)
        y = self.labels[idx]
        n = idx.numel()
        zr, sr, fl = affine if affine else (0.0, 0.0, False)
        # Noise is drawn deterministically per call position rather than per
        # index (cheap); the class template carries the learnable signal.
        # Sampled directly on the device — the training loop never touches host.
        if self.device.type == "cuda":
            if self.dtype == torch.bfloat16:
                # single fused kernel: template gather (affine-sampled when
                # augmenting) + counter-hash normal noise + sigmoid + bf16
                import hefl
                self._synth_ctr = getattr(self, "_synth_ctr", 0) + 1
                return hefl.load_extension().synth_batch(
                    self.templates, y.contiguous(),
                    self.seed * 0x10001 + self._synth_ctr,
                    zoom=zr, shear=sr, flip=int(bool(fl))), y
            if not hasattr(self, "_gen"):
                self._gen = torch.Generator(device=self.device)
                self._gen.manual_seed(self.seed)
            noise = torch.randn(n, self.H, self.W, self.C, generator=self._gen,
                                device=self.device)
        else:
            g = torch.Generator(device="cpu").manual_seed(self.seed)
            noise = torch.randn(n, self.H, self.W, self.C, generator=g)
        tmpl = self.templates[y]
        if affine and (zr or sr or fl):
            tmpl = self._affine_sample(tmpl, zr, sr, fl)
        x = 0.6 * tmpl + 0.4 * noise
        x = torch.sigmoid(x)  # [0, 1] like rescale=1/255 images
        return x.to(self.dtype), y

    def _affine_sample(self, tmpl: torch.Tensor, zr: float, sr: float,
                       fl: bool) -> torch.Tensor:
        """Per-sample zoom/shear/h-flip via border-clamped bilinear sampling
        (torch reference path; the GPU bf16 path fuses the same transform
        into the synth kernel)."""
        self._aug_ctr = getattr(self, "_aug_ctr", 0) + 1
        g = torch.Generator(device="cpu").manual_seed(
            self.seed * 31 + self._aug_ctr)
        return affine_sample(tmpl, zr, sr, fl, g)


def affine_sample(x: torch.Tensor, zr: float, sr: float, fl: bool,
                  g: torch.Generator) -> torch.Tensor:
    """Per-sample random zoom/shear/h-flip of NHWC images via inverse-map
    border-clamped bilinear sampling (the reference ImageDataGenerator
    transform set, FLPyfhelin.py:80-86). Shared by the synthetic dataset's
    CPU path and the file-backed dataset."""
    import torch.nn.functional as F
    n, H, W = x.shape[0], x.shape[1], x.shape[2]
    zoom = 1 + zr * (2 * torch.rand(n, generator=g) - 1)
    shear = sr * (2 * torch.rand(n, generator=g) - 1)
    sf = torch.where(torch.rand(n, generator=g) < 0.5, -1.0, 1.0) \
        if fl else torch.ones(n)
    # inverse map (output -> source), about the image center; grid_sample
    # normalized coords with align_corners=True match the (dim-1) scaling
    hh = torch.arange(H, dtype=torch.float32)
    ww = torch.arange(W, dtype=torch.float32)
    cy, cx = 0.5 * (H - 1), 0.5 * (W - 1)
    dy = (hh - cy).view(1, H, 1)
    dx = (ww - cx).view(1, 1, W)
    sy = cy + dy / zoom.view(-1, 1, 1) + torch.zeros(1, 1, W)
    sx = (cx + dx * (sf / zoom).view(-1, 1, 1)
          + shear.view(-1, 1, 1) * dy)
    grid = torch.stack([2 * sx / (W - 1) - 1, 2 * sy / (H - 1) - 1],
                       dim=-1).to(x.device, torch.float32)
    out = F.grid_sample(x.float().permute(0, 3, 1, 2), grid, mode="bilinear",
                        padding_mode="border", align_corners=True)
    return out.permute(0, 2, 3, 1).to(x.dtype)


def make_client_loader(ds: SyntheticMedicalImages, client: int, n_clients: int,
                       batch_size: int, seed: int = 0,
                       shuffle: bool = True) -> "ClientLoader":
    from .shard import shard_indices
    idx = shard_indices(ds.n_samples * n_clients, client, n_clients)
    # Clamp to this dataset's local size: callers may build a per-client ds
    idx = idx[idx < ds.n_samples] if idx.numel() and idx.max() >= ds.n_samples else idx
    return ClientLoader(ds, idx, batch_size, seed=seed, shuffle=shuffle)


class ClientLoader:
    """Minimal epoch iterator over a shard (drop_last=False, like Keras)."""

    def __init__(self, ds: SyntheticMedicalImages, indices: torch.Tensor,
                 batch_size: int, seed: int = 0, shuffle: bool = True,
                 augment=None, affine=None):
        self.ds = ds
        self.indices = indices.clone()
        self.batch_size = int(batch_size)
        self.seed = int(seed)
        self.shuffle = shuffle
        self.augment = augment  # callable(x) -> x, e.g. pipeline.hflip_augment
        self.affine = affine    # (zoom, shear, hflip) in-generator transform
        self._epoch = 0

    def __len__(self) -> int:
        return (self.indices.numel() + self.batch_size - 1) // self.batch_size

    def epoch_order(self) -> torch.Tensor:
        """Consume one epoch's (shuffled) sample order — used by the
        epoch-graph path, which stages the whole epoch at once."""
        order = self.indices
        if self.shuffle:
            g = torch.Generator().manual_seed(self.seed + self._epoch)
            order = order[torch.randperm(order.numel(), generator=g)]
        self._epoch += 1
        return order

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        order = self.epoch_order()
        for i in range(0, order.numel(), self.batch_size):
            x, y = self.ds.batch(order[i:i + self.batch_size],
                                 affine=self.affine)
            if self.augment is not None:
                x = self.augment(x)
            yield x, y
