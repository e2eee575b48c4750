"""Data pipelines — API parity with the reference's Keras generators.

Reference: get_test_data (FLPyfhelin.py:57-71) builds a rescale-only
categorical test iterator; get_train_data (:73-114) slices client shard
i*ratio:(i+1)*ratio, applies augmentation (shear 0.2, zoom 0.2, horizontal
flip) and splits 10% validation. Here the dataset is synthetic and already
[0,1]-scaled (SURVEY.md data row); augmentation keeps the horizontal-flip
hook (shear/zoom are file-image ops without an analog on synthetic blobs —
the hook point `ClientLoader(augment=...)` is where they would go).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from .shard import shard_indices
from .synthetic import ClientLoader, SyntheticMedicalImages


def hflip_augment(x: torch.Tensor) -> torch.Tensor:
    """Random horizontal flip per sample (reference: horizontal_flip=True,
    FLPyfhelin.py:85)."""
    n = x.shape[0]
    mask = torch.rand(n, device=x.device) < 0.5
    flipped = x.flip(2)  # NHWC: W axis
    return torch.where(mask.view(-1, 1, 1, 1), flipped, x)


def get_test_data(ds: SyntheticMedicalImages, batch_size: int = 32) -> ClientLoader:
    """No shuffle, no augmentation (reference get_test_data, FLPyfhelin.py:57-71)."""
    idx = torch.arange(ds.n_samples)
    return ClientLoader(ds, idx, batch_size, shuffle=False)


def get_train_data(ds: SyntheticMedicalImages, client: int, n_clients: int,
                   batch_size: int = 32, val_frac: float = 0.1,
                   seed: int = 0, augment: bool = True,
                   affine=None) -> Tuple[ClientLoader, Optional[ClientLoader]]:
    """Client shard -> (train loader, val loader).

    Shard semantics are the reference's contiguous equal slices
    (FLPyfhelin.py:75-78); validation is the trailing val_frac of the shard
    (reference: validation_split=0.1, :88-99). `affine` = (zoom, shear,
    hflip) runs the full in-generator transform (the reference's shear 0.2
    / zoom 0.2 / h-flip set, :80-86); `augment` alone keeps the post-batch
    h-flip hook.
    """
    idx = shard_indices(ds.n_samples, client, n_clients)
    n_val = int(idx.numel() * val_frac)
    train_idx = idx[: idx.numel() - n_val]
    val_idx = idx[idx.numel() - n_val:]
    aug = hflip_augment if (augment and affine is None) else None
    train = ClientLoader(ds, train_idx, batch_size, seed=seed, augment=aug,
                         affine=affine)
    val = (ClientLoader(ds, val_idx, batch_size, shuffle=False)
           if n_val else None)
    return train, val
