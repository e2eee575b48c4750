"""Configuration for hefl.

The reference keeps config as module constants (FLPyfhelin.py:31-36) plus
notebook-cell variables; here it is explicit dataclasses with the BASELINE.json
north-star configs as named presets.
"""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Optional, Tuple


@dataclass
class HEConfig:
    """CKKS parameters.

    `m` is the polynomial modulus degree (ring dimension n) — the Pyfhel-2.3.1
    parameter name is kept (reference FLPyfhelin.py:332 `contextGen(p=..., m=...)`).
    RNS limb primes are < 2**60 so a lazy int64 sum over <=8 clients cannot
    overflow during the RCCL all-reduce (collectives row).
    """
    m: int = 8192                       # ring degree n (power of two)
    scale_bits: int = 40                # CKKS scale Delta = 2**scale_bits
    q_bits: Tuple[int, ...] = (60, 40)  # RNS prime chain bit sizes (q0 first)
    sec: int = 128                      # advisory security level (API compat)
    p: int = 65537                      # accepted for Pyfhel API compat; unused by CKKS
    seed: Optional[int] = None          # deterministic keygen when set

    @property
    def n(self) -> int:
        return self.m

    @property
    def slots(self) -> int:
        return self.m // 2

    def __post_init__(self):
        if self.m & (self.m - 1) or self.m < 16:
            raise ValueError(f"m must be a power of two >= 16, got {self.m}")
        for b in self.q_bits:
            if not (20 <= b <= 60):
                raise ValueError(f"q limb bits must be in [20, 60] (lazy-allreduce bound), got {b}")


@dataclass
class ModelConfig:
    name: str = "cnn2"                    # cnn2 | lenet5 | refcnn6 | cnn4 | resnet18
    in_shape: Tuple[int, int, int] = (28, 28, 1)  # H, W, C (NHWC)
    n_classes: int = 10


@dataclass
class TrainConfig:
    # Reference defaults: INIT_LR=1e-3, EPOCHS=10, BS=32 (FLPyfhelin.py:31-33)
    lr: float = 1e-3
    lr_decay: float = 1e-4                # Adam 'decay' as in Keras (lr_t = lr/(1+decay*step))
    batch_size: int = 32
    local_epochs: int = 10
    beta1: float = 0.9
    beta2: float = 0.999
    eps: float = 1e-7                     # Keras Adam default epsilon
    dtype: str = "bf16"                   # compute dtype on GPU; fp32 on CPU
    hip_graphs: bool = True               # capture the train step in a hipGraph (GPU)


@dataclass
class FLConfig:
    n_clients: int = 2
    samples_per_client: int = 720         # reference: 800/client, 720 train / 80 val
    val_samples_per_client: int = 80
    test_samples: int = 400
    encrypted: bool = True                # False -> plaintext FedAvg (config #1)
    denom_mode: str = "plain"             # "encrypted" -> ct x ct + relin in aggregation (config #3)
    # training-data augmentation: "none" | "hflip" | "full" ("full" = the
    # reference's ImageDataGenerator set: shear 0.2, zoom 0.2, h-flip —
    # FLPyfhelin.py:80-86 — applied in the data generator)
    augment: str = "none"
    seed: int = 1234


@dataclass
class RunConfig:
    model: ModelConfig = field(default_factory=ModelConfig)
    train: TrainConfig = field(default_factory=TrainConfig)
    fl: FLConfig = field(default_factory=FLConfig)
    he: HEConfig = field(default_factory=HEConfig)

    def replace(self, **kw) -> "RunConfig":
        return dataclasses.replace(self, **kw)


def preset(name: str) -> RunConfig:
    """Named presets mirroring BASELINE.json `configs` #1-#5."""
    if name in ("plumbing", "config1"):
        # 2-client, 2-conv CNN, 28x28x1, plaintext aggregation (CPU/gloo-able)
        return RunConfig(
            model=ModelConfig("cnn2", (28, 28, 1), 10),
            fl=FLConfig(n_clients=2, encrypted=False),
            he=HEConfig(m=8192),
        )
    if name in ("headline", "config2", "cnn2-ckks"):
        # 8-client 2-conv CNN on 28x28x1, CKKS n=2^13 encrypted FedAvg
        return RunConfig(
            model=ModelConfig("cnn2", (28, 28, 1), 10),
            fl=FLConfig(n_clients=8, encrypted=True),
            he=HEConfig(m=8192, scale_bits=40, q_bits=(60, 40)),
        )
    if name in ("config3", "lenet5-ckks"):
        # 8-client LeNet-5 on 32x32x3, CKKS n=2^14 with homomorphic mult + rescale
        return RunConfig(
            model=ModelConfig("lenet5", (32, 32, 3), 10),
            fl=FLConfig(n_clients=8, encrypted=True, denom_mode="encrypted"),
            he=HEConfig(m=16384, scale_bits=40, q_bits=(60, 40, 40)),
        )
    if name in ("config4", "cnn4-xray"):
        # 8-client 4-conv CNN on 224x224x1 chest-X-ray-shaped synthetic, n=2^14
        return RunConfig(
            model=ModelConfig("cnn4", (224, 224, 1), 2),
            fl=FLConfig(n_clients=8, encrypted=True, samples_per_client=720),
            he=HEConfig(m=16384, scale_bits=40, q_bits=(60, 40)),
        )
    if name in ("config5", "resnet18-ckks"):
        # 8-client ResNet-18 on 128x128x3, CKKS n=2^15 deep RNS chain
        return RunConfig(
            model=ModelConfig("resnet18", (128, 128, 3), 10),
            fl=FLConfig(n_clients=8, encrypted=True, samples_per_client=720),
            he=HEConfig(m=32768, scale_bits=40, q_bits=(60, 40, 40, 40)),
        )
    if name in ("reference", "refcnn6"):
        # The reference's own config: 2 clients, 6-conv CNN, 256x256x3,
        # 2 classes, full augmentation (FLPyfhelin.py:80-86,118-146)
        return RunConfig(
            model=ModelConfig("refcnn6", (256, 256, 3), 2),
            fl=FLConfig(n_clients=2, encrypted=True, augment="full"),
            he=HEConfig(m=8192),
        )
    raise KeyError(f"unknown preset: {name}")
