"""Encrypted-weight file export/import — API parity with the reference's
pickle transport (FLPyfhelin.py:200-249 encrypt_export_weights/export_weights,
:303-328 import_encrypted_weights).

The live FL path never touches disk (ciphertexts stay HBM-resident and move
by RCCL all-reduce, hefl/fl/secure.py); these functions exist for the
reference's file-based workflow: key-separated hand-off of encrypted weights
between processes/machines, and durable encrypted round dumps. Layout is the
reference's dict shape {'key': <HE>, 'val': {'c_<i>_<j>': <ciphertext>}} with
slot-packed CtxtTensors per parameter instead of per-scalar PyCtxt arrays.
Timing prints keep the reference's labels (FLPyfhelin.py:224,239,248,327).
"""
from __future__ import annotations

import os
import pickle
import time
from typing import Dict

import torch

from ..he.ckks import CtxtTensor
from ..he.pyfhel_compat import Pyfhel


def encrypt_model_weights(he: Pyfhel, model: torch.nn.Module) -> Dict[str, dict]:
    """Encrypt every parameter tensor: key 'c_<param index>_<0>' mirrors the
    reference's 'c_<layer>_<tensor>' naming (FLPyfhelin.py:221)."""
    start = time.time()
    out: Dict[str, dict] = {}
    for i, p in enumerate(model.parameters()):
        ct = he.encrypt_tensor(p.detach().float().reshape(-1))
        out[f"c_{i}_0"] = {
            "data": ct.data.cpu(),
            "scale": ct.scale,
            "count": ct.count,
            "shape": tuple(p.shape),
        }
    print("Time to encrypt weights:", time.time() - start)
    return out


def export_weights(file_name: str, weights: dict, he: Pyfhel) -> None:
    """Reference export_weights (FLPyfhelin.py:230-240): one pickle with the
    public material and the ciphertext dict."""
    start = time.time()
    os.makedirs(os.path.dirname(file_name) or ".", exist_ok=True)
    payload = {
        "key": {"context": he.to_bytes_context(),
                "public_key": he.to_bytes_publicKey()},
        "val": weights,
    }
    with open(file_name, "wb") as f:
        pickle.dump(payload, f, protocol=pickle.HIGHEST_PROTOCOL)
    print("Time to export weights to pickle:", time.time() - start)


def encrypt_export_weights(he: Pyfhel, model: torch.nn.Module, client_id: int,
                           directory: str = "weights") -> str:
    """Reference encrypt_export_weights (FLPyfhelin.py:200-228)."""
    weights = encrypt_model_weights(he, model)
    path = os.path.join(directory, f"client_{client_id + 1}.pickle")
    export_weights(path, weights, he)
    print("Weights exported: Client", client_id + 1)
    return path


def import_encrypted_weights(file_name: str, device: str = "cpu"):
    """Reference import_encrypted_weights (FLPyfhelin.py:303-328): returns
    (he, {name: CtxtTensor}). No context re-attach hack needed — the context
    params ride in the file header."""
    start = time.time()
    with open(file_name, "rb") as f:
        payload = pickle.load(f)
    he = Pyfhel()
    he._device = device
    he.from_bytes_context(payload["key"]["context"])
    he.from_bytes_publicKey(payload["key"]["public_key"])
    val = {}
    for name, d in payload["val"].items():
        val[name] = CtxtTensor(d["data"].to(device), d["scale"], d["count"])
    print("Time to import:", time.time() - start)
    return he, val


def decrypt_into_model(he: Pyfhel, val: Dict[str, CtxtTensor],
                       model: torch.nn.Module) -> torch.nn.Module:
    """Reference decrypt_import_weights tail (FLPyfhelin.py:263-281)."""
    start = time.time()
    with torch.no_grad():
        for i, p in enumerate(model.parameters()):
            d = val[f"c_{i}_0"]
            ct = d if isinstance(d, CtxtTensor) else CtxtTensor(
                d["data"], d["scale"], d["count"])
            vec = he.decrypt_tensor(ct)
            p.copy_(vec.reshape(p.shape).to(p.dtype))
    print("Time to decrypt:", time.time() - start)
    return model


def export_plain_weights(model: torch.nn.Module, file_name: str = "plainweights.pickle") -> str:
    """Plaintext weight export for size/time comparison — API parity with
    the reference's notebook cell 6 (load_weights + export_weights of
    unencrypted arrays to plainweights.pickle)."""
    start = time.time()
    os.makedirs(os.path.dirname(file_name) or ".", exist_ok=True)
    val = {f"c_{i}_0": p.detach().float().cpu().numpy()
           for i, p in enumerate(model.parameters())}
    with open(file_name, "wb") as f:
        pickle.dump({"key": None, "val": val}, f,
                    protocol=pickle.HIGHEST_PROTOCOL)
    print("Time to export weights to pickle:", time.time() - start)
    return file_name
