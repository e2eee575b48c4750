"""File-backed dataset: `folder/<class>/*.npy` HWC arrays.

The reference trains on image folders through Keras ImageDataGenerator
(`flow_from_dataframe` over prep_df's (Path, Label) frame,
FLPyfhelin.py:38-55,57-114). The north-star benchmarks use synthetic data
(no network for datasets), but a migrating user with real data gets the
same folder/<class> layout here — as .npy arrays, since no image-decode
library ships in the offline environment. Presents the same
`batch(indices, affine=...)` surface as SyntheticMedicalImages, so
ClientLoader / get_train_data / get_test_data work unchanged.
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from .shard import prep_df
from .synthetic import affine_sample


class FileImageDataset:
    """Images from `folder/<class>/*.npy` (HWC float or uint8 arrays).

    uint8 arrays are rescaled by 1/255 like the reference's
    ImageDataGenerator(rescale=1./255) (FLPyfhelin.py:59); float arrays are
    taken as-is. The whole set is materialized once (`cache=True`, default)
    — medical-image folders at the reference's scale (~1200 images of
    256x256x3) are ~300 MB, trivial beside 288 GB of HBM.
    """

    def __init__(self, folder: str, device: str = "cpu",
                 dtype: torch.dtype = torch.float32, cache: bool = True,
                 seed: int = 0):
        self.df = prep_df(folder, shuffle=False)
        if len(self.df) == 0:
            raise ValueError(f"no files under {folder}/<class>/")
        self.classes = sorted(self.df["Label"].unique())
        self.class_to_idx = {c: i for i, c in enumerate(self.classes)}
        self.n_classes = len(self.classes)
        self.n_samples = len(self.df)
        self.device = torch.device(device)
        self.dtype = dtype
        self.seed = int(seed)
        self.labels = torch.tensor(
            [self.class_to_idx[l] for l in self.df["Label"]],
            dtype=torch.int64, device=self.device)
        self._paths = list(self.df["Path"])
        self._cache: Optional[torch.Tensor] = None
        if cache:
            imgs = [self._load(p) for p in self._paths]
            shapes = {tuple(i.shape) for i in imgs}
            if len(shapes) != 1:
                raise ValueError(f"mixed image shapes in {folder}: {shapes}")
            self._cache = torch.stack(imgs).to(self.device)
        h, w, c = (self._cache.shape[1:] if self._cache is not None
                   else self._load(self._paths[0]).shape)
        self.H, self.W, self.C = int(h), int(w), int(c)

    @staticmethod
    def _load(path: str) -> torch.Tensor:
        a = np.load(path, allow_pickle=False)
        if a.ndim == 2:
            a = a[..., None]
        if a.dtype == np.uint8:
            a = a.astype(np.float32) / 255.0  # reference rescale=1/255
        return torch.from_numpy(np.ascontiguousarray(a, dtype=np.float32))

    def batch(self, indices: torch.Tensor,
              affine=None) -> Tuple[torch.Tensor, torch.Tensor]:
        idx = indices.to(self.device)
        y = self.labels[idx]
        if self._cache is not None:
            x = self._cache.index_select(0, idx)
        else:
            x = torch.stack([self._load(self._paths[int(i)])
                             for i in indices]).to(self.device)
        if affine is not None and (affine[0] or affine[1] or affine[2]):
            zr, sr, fl = affine
            self._aug_ctr = getattr(self, "_aug_ctr", 0) + 1
            g = torch.Generator(device="cpu").manual_seed(
                self.seed * 31 + self._aug_ctr)
            x = affine_sample(x, zr, sr, fl, g)
        return x.to(self.dtype), y
