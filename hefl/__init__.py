"""hefl — MI355X-native privacy-preserving federated CNN training.

A from-scratch AMD MI355X (gfx950 / CDNA4) framework with the capabilities of
the reference `FebriantiW/Homomorphic-Encryption-and-Federated-Learning-based-
Privacy-Preserving-CNN-Training-` (Pyfhel/SEAL + Keras notebook):

- Local CNN training per federated client (1 MI355X GPU = 1 client) through
  hand-written HIP/CDNA4 kernels (MFMA implicit-GEMM conv, pool, fused
  softmax-CE, fused Adam) — replacing TF/Keras (reference FLPyfhelin.py:118-146).
- A from-scratch CKKS homomorphic-encryption layer in HIP (batched NTT/INTT
  over RNS limbs, encode/encrypt, homomorphic add / plain-mult, rescale) —
  replacing Pyfhel 2.3.1 -> SEAL 2.3 BFV (reference FLPyfhelin.py:330-344).
- Encrypted FedAvg as an RCCL all-reduce over xGMI on raw RNS coefficient
  tensors — replacing pickle files on disk (reference FLPyfhelin.py:230-240).
- A Pyfhel-2.3.1-shaped compatibility API (`contextGen(m=...)`, `encryptFrac`,
  `decryptFrac`, `to_bytes_*` / `from_bytes_*`, picklable ciphertexts).
"""

__version__ = "0.1.0"

from . import config  # noqa: F401

_C = None
_C_IMPORT_ERROR = None


def load_extension():
    """Import the in-tree HIP extension (hefl._C). Returns the module.

    On a GPU box the extension is REQUIRED for every hefl op that touches a
    CUDA tensor: ops raise RuntimeError rather than silently falling back to
    eager PyTorch.
    """
    global _C, _C_IMPORT_ERROR
    if _C is not None:
        return _C
    try:
        # NOTE: must be importlib, not `from . import _C` — the package-level
        # `_C = None` attribute above would shadow the submodule import.
        import importlib
        _C = importlib.import_module("hefl._C")
    except ImportError as e:  # pragma: no cover
        _C_IMPORT_ERROR = e
        raise RuntimeError(
            "hefl._C HIP extension is not built. Run "
            "`python setup.py build_ext --inplace` (cross-compiles for gfx950). "
            f"Original error: {e}"
        ) from e
    return _C


def has_extension() -> bool:
    try:
        load_extension()
        return True
    except RuntimeError:
        return False
