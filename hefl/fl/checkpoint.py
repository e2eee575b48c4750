"""Checkpoint / resume.

Keeps the reference's save_weights/load_weights(ind) API shape (object-array
.npy of per-parameter tensors, FLPyfhelin.py:149-159) plus a single
round-state checkpoint format (model + optimizer + HE keys + round counter,
SURVEY.md section 5 checkpoint row) that gives round-granularity resume —
the reference gets this only implicitly from its scattered pickle files.
"""
from __future__ import annotations

import os
from typing import Optional

import numpy as np
import torch


def save_weights(model: torch.nn.Module, ind: str, directory: str = "weights"):
    """Reference-shaped: weights/weights<ind>.npy, object array of per-param
    ndarrays in parameter order (FLPyfhelin.py:149-153)."""
    os.makedirs(directory, exist_ok=True)
    arrs = np.empty(len(list(model.parameters())), dtype=object)
    for i, p in enumerate(model.parameters()):
        arrs[i] = p.detach().float().cpu().numpy()
    np.save(os.path.join(directory, f"weights{ind}.npy"), arrs,
            allow_pickle=True)


def load_weights(model: torch.nn.Module, ind: str, directory: str = "weights"):
    """Inverse of save_weights: loads into an existing model in place
    (the reference rebuilds the model first, FLPyfhelin.py:155-159)."""
    arrs = np.load(os.path.join(directory, f"weights{ind}.npy"),
                   allow_pickle=True)
    with torch.no_grad():
        for p, a in zip(model.parameters(), arrs):
            p.copy_(torch.from_numpy(np.ascontiguousarray(a)).to(p.dtype))
    return model


def save_model_weights(model: torch.nn.Module, path: str):
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    torch.save({k: v.cpu() for k, v in model.state_dict().items()}, path)


def load_model_weights(model: torch.nn.Module, path: str):
    model.load_state_dict(torch.load(path, weights_only=True))
    return model


def save_round_state(path: str, model: torch.nn.Module, optimizer,
                     round_idx: int, he=None, extra: Optional[dict] = None,
                     private_path: Optional[str] = None):
    """One durable artifact per round: enough to resume the federation.

    Key separation mirrors the reference's publickey/privatekey split
    (notebook cell 1, FLPyfhelin.py:251-261): the round file carries only
    PUBLIC HE material (context + pk); the secret key goes to a separate
    private artifact (default `<path>.private`) that only the decrypting
    party needs — so a round checkpoint can be shared with the aggregation
    server without leaking sk."""
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    state = {
        "round": round_idx,
        "model": {k: v.cpu() for k, v in model.state_dict().items()},
        "optimizer": {
            "step": optimizer.state_dict()["step"],
            "lr": optimizer.lr,
            "decay": optimizer.decay,
            "m": [t.cpu() for t in optimizer.m],
            "v": [t.cpu() for t in optimizer.v],
        },
        "extra": extra or {},
    }
    if he is not None:  # hefl.he.Pyfhel facade
        state["he"] = {
            "context": he.to_bytes_context(),
            "public_key": he.to_bytes_publicKey(),
        }
        if he._sk is not None:
            torch.save({"context": he.to_bytes_context(),
                        "secret_key": he.to_bytes_secretKey()},
                       private_path or path + ".private")
    torch.save(state, path)


def load_round_state(path: str, model: torch.nn.Module, optimizer, he=None,
                     private_path: Optional[str] = None):
    state = torch.load(path, weights_only=False)
    model.load_state_dict(state["model"])
    opt = state["optimizer"]
    # go through the optimizer API: a graphed client's captured kernels read
    # DEVICE-side step/lr buffers, which raw attribute writes would miss
    if hasattr(optimizer, "set_step"):
        optimizer.set_step(opt["step"])
        optimizer.set_lr(opt["lr"])
        optimizer.decay = opt["decay"]
        if hasattr(optimizer, "_hyper"):
            optimizer._hyper[1] = opt["decay"]
    else:
        optimizer.step_count = opt["step"]
        optimizer.lr = opt["lr"]
        optimizer.decay = opt["decay"]
    for dst, src in zip(optimizer.m, opt["m"]):
        dst.copy_(src.to(dst.device))
    for dst, src in zip(optimizer.v, opt["v"]):
        dst.copy_(src.to(dst.device))
    if he is not None and "he" in state:
        he.from_bytes_context(state["he"]["context"])
        he.from_bytes_publicKey(state["he"]["public_key"])
        if "secret_key" in state["he"]:  # legacy bundled-sk round files
            he.from_bytes_secretKey(state["he"]["secret_key"])
        else:
            sk_file = private_path or path + ".private"
            if os.path.exists(sk_file):
                priv = torch.load(sk_file, weights_only=False)
                he.from_bytes_secretKey(priv["secret_key"])
    return state["round"], state.get("extra", {})
