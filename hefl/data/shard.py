"""Client sharding + dataset indexing.

Reproduces the reference's sharding semantics: equal contiguous shards,
ratio = len // num_clients, shard i = df[i*ratio : (i+1)*ratio]
(FLPyfhelin.py:75-78). Tail samples beyond n_clients*ratio are dropped,
exactly as the reference's slicing does.
"""
from __future__ import annotations

import os

import torch


def shard_indices(n_total: int, client: int, n_clients: int) -> torch.Tensor:
    """Equal contiguous shard of [0, n_total) for `client` (0-based)."""
    if not (0 <= client < n_clients):
        raise ValueError(f"client {client} out of range [0, {n_clients})")
    ratio = n_total // n_clients
    return torch.arange(client * ratio, (client + 1) * ratio, dtype=torch.long)


def prep_df(folder: str, shuffle: bool = True, seed: int = 42):
    """Index a directory tree `folder/<class>/*` into a DataFrame of
    (Path, Label) — API parity with the reference's prep_df
    (FLPyfhelin.py:38-55). Requires pandas (available offline)."""
    import pandas as pd

    rows = []
    for label in sorted(os.listdir(folder)):
        cls_dir = os.path.join(folder, label)
        if not os.path.isdir(cls_dir):
            continue
        for f in sorted(os.listdir(cls_dir)):
            rows.append((os.path.join(cls_dir, f), label))
    df = pd.DataFrame(rows, columns=["Path", "Label"])
    if shuffle:
        df = df.sample(frac=1.0, random_state=seed).reset_index(drop=True)
    return df
