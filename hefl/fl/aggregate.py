"""FedAvg aggregation over torch.distributed.

Plaintext path (BASELINE.json config #1): one all-reduce(SUM) on the flat
fp32 weight vector, divided by world size — the collective realization of the
reference's per-scalar loop `dct[key] = enc[key] + dct[key]; dct[key] *= 1/n`
(FLPyfhelin.py:366-390) without HE.

Encrypted path: see hefl.fl.secure (CKKS ciphertext all-reduce with lazy
int64 reduction; limbs < 2**60 so 8 summands cannot overflow int64 —
SURVEY.md section 5 collectives row).
"""
from __future__ import annotations

import torch
import torch.distributed as dist


def plaintext_fedavg(vec: torch.Tensor) -> torch.Tensor:
    """All-reduce-average a flat fp32 weight vector across all ranks."""
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(vec, op=dist.ReduceOp.SUM)
        vec /= dist.get_world_size()
    return vec
