"""Flat fp32 weight vector <-> model parameters.

The aggregation unit is the model's full fp32 parameter vector (the reference
aggregates per-layer weight tensors, FLPyfhelin.py:200-228; flattening them
into one vector lets encryption slot-pack and the all-reduce run as one
bucketed collective). Order = model.parameters() order, deterministic.
"""
from __future__ import annotations

from typing import List

import torch


def _agg_tensors(model: torch.nn.Module):
    """Parameters + floating-point buffers (BN running stats must be averaged
    across clients too), in deterministic module order."""
    for p in model.parameters():
        yield p
    for b in model.buffers():
        if b.is_floating_point():
            yield b


def flat_params(model: torch.nn.Module) -> torch.Tensor:
    with torch.no_grad():
        return torch.cat([p.detach().float().reshape(-1)
                          for p in _agg_tensors(model)])


def load_flat_params(model: torch.nn.Module, vec: torch.Tensor) -> None:
    with torch.no_grad():
        off = 0
        for p in _agg_tensors(model):
            n = p.numel()
            p.copy_(vec[off:off + n].reshape(p.shape).to(p.dtype))
            off += n
        if off != vec.numel():
            raise ValueError(f"vector length {vec.numel()} != param count {off}")


def param_shapes(model: torch.nn.Module) -> List[torch.Size]:
    return [p.shape for p in model.parameters()]
