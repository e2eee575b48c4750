"""Pyfhel-2.3.1-shaped compatibility API over the hefl CKKS engine.

The reference drives all HE through Pyfhel 2.3.1 (README.md:7 pins the
version; FLPyfhelin.py:330-344 contextGen/keyGen, :217 encryptFrac, :295
decryptFrac, :337-338/:352-353 to_bytes_/from_bytes_ context/publicKey/
secretKey, :381 PyCtxt.__add__, :385 PyCtxt.__mul__(float), pickled
ciphertexts with a context re-attach at :321).

This module keeps that exact surface — same names, same parameter names
(`m`, not `n`; `p` accepted and ignored by CKKS) — while the engine
underneath is the from-scratch RNS-CKKS of hefl.he.ckks (slot-packing,
HIP kernels on MI355X). Deliberate deltas (SURVEY.md §7 quirks catalog):
`relinKeyGen` works instead of raising NameError (reference
FLPyfhelin.py:363 is dead+buggy), and a pickled PyCtxt round-trips
without a live context (`_pyfhel` re-attach still supported, no longer
required).
"""
from __future__ import annotations

import io
import struct
from typing import Optional

import numpy as np
import torch

from ..config import HEConfig
from .ckks import CKKSContext, Ciphertext, CtxtTensor, KeyPair

_MAGIC = b"HEFL"
_VERSION = 1


def _pack_header(kind: bytes, cfg: HEConfig) -> bytes:
    return _MAGIC + struct.pack(
        "<H4sIIIH", _VERSION, kind, cfg.m, cfg.scale_bits, cfg.sec,
        len(cfg.q_bits)) + struct.pack(f"<{len(cfg.q_bits)}I", *cfg.q_bits)


def _unpack_header(buf: bytes, expect: bytes):
    if buf[:4] != _MAGIC:
        raise ValueError("not a hefl HE byte stream")
    ver, kind, m, scale_bits, sec, nq = struct.unpack("<H4sIIIH", buf[4:24])
    if ver != _VERSION:
        raise ValueError(f"unsupported version {ver}")
    if kind != expect:
        raise ValueError(f"expected {expect!r} stream, got {kind!r}")
    q_bits = struct.unpack(f"<{nq}I", buf[24:24 + 4 * nq])
    cfg = HEConfig(m=m, scale_bits=scale_bits, sec=sec, q_bits=tuple(q_bits))
    return cfg, 24 + 4 * nq


class PyCtxt:
    """A single CKKS ciphertext with Pyfhel-2.3.1 PyCtxt semantics.

    Supports `ct + ct`, `ct + 0` (the reference seeds its accumulator with
    int zeros, FLPyfhelin.py:380-381), and `ct * float` (plaintext-scalar
    multiply, :385). Picklable standalone; `_pyfhel` attribute kept for
    reference-style re-attach (FLPyfhelin.py:321) but a fresh context is
    rebuilt from the embedded params when absent.
    """

    def __init__(self, ct: Ciphertext, pyfhel: "Pyfhel"):
        self._ct = ct
        self._pyfhel = pyfhel

    # ----- arithmetic (the ops the reference exercises) -----
    def _he(self) -> "Pyfhel":
        if self._pyfhel is None:
            raise ValueError("PyCtxt has no attached Pyfhel context")
        return self._pyfhel

    def __add__(self, other):
        if isinstance(other, (int, float)) and other == 0:
            # clone, not self: the reference seeds accumulators with int 0
            # (FLPyfhelin.py:380-381) — returning self would alias the
            # accumulator to the client's ciphertext, so a later in-place
            # mutation of the sum would corrupt the client's upload
            return PyCtxt(self._ct.clone(), self._pyfhel)
        if isinstance(other, PyCtxt):
            he = self._he()
            return PyCtxt(he._ctx.add(self._ct, other._ct), he)
        return NotImplemented

    __radd__ = __add__

    def __mul__(self, other):
        if isinstance(other, (int, float)):
            he = self._he()
            out = he._ctx.mul_scalar(self._ct, float(other))
            if out.level > 1:
                out = he._ctx.rescale(out)
            return PyCtxt(out, he)
        return NotImplemented

    __rmul__ = __mul__

    # ----- pickling: standalone, context params embedded -----
    def __getstate__(self):
        he = self._pyfhel
        return {
            "cfg": he._ctx.cfg if he is not None else None,
            "data": self._ct.data.cpu().numpy(),
            "scale": self._ct.scale,
        }

    def __setstate__(self, state):
        self._ct = Ciphertext(torch.from_numpy(state["data"]), state["scale"])
        self._pyfhel = Pyfhel() if state["cfg"] is None else None
        if state["cfg"] is not None:
            he = Pyfhel()
            he.contextGen_cfg(state["cfg"])
            self._pyfhel = he


class Pyfhel:
    """Pyfhel-2.3.1-compatible facade (reference API at FLPyfhelin.py:330-364).

    `contextGen(p=..., m=..., sec=...)` keeps the Pyfhel signature: `m` is
    the ring degree, `p` is accepted for BFV compat and ignored by CKKS,
    `sec` is advisory. Extra kwargs (scale_bits, q_bits, device) configure
    the CKKS engine.
    """

    def __init__(self):
        self._ctx: Optional[CKKSContext] = None
        self._keys: Optional[KeyPair] = None
        self._pk: Optional[torch.Tensor] = None
        self._sk: Optional[torch.Tensor] = None
        self._device = "cpu"

    # ----- context -----
    def contextGen(self, p: int = 65537, m: int = 2048, sec: int = 128,
                   base: int = 2, flagBatching: bool = False,
                   scale_bits: int = 40, q_bits=(60, 40),
                   seed: Optional[int] = None, device: str = "cpu"):
        cfg = HEConfig(m=m, scale_bits=scale_bits, q_bits=tuple(q_bits),
                       sec=sec, p=p, seed=seed)
        self.contextGen_cfg(cfg, device=device)

    def contextGen_cfg(self, cfg: HEConfig, device: str = "cpu"):
        self._device = device
        self._ctx = CKKSContext(cfg, device=device)

    @property
    def context(self) -> CKKSContext:
        if self._ctx is None:
            raise ValueError("contextGen has not been called")
        return self._ctx

    def __repr__(self):
        if self._ctx is None:
            return "<Pyfhel (no context)>"
        c = self._ctx.cfg
        # mirrors the stored notebook repr shape: contx(p=..., m=..., ...)
        return (f"<Pyfhel CKKS contx(p={c.p}, m={c.m}, base=2, sec={c.sec}, "
                f"scale=2^{c.scale_bits}, qbits={list(c.q_bits)}, batch=True)>")

    # ----- keys -----
    def keyGen(self):
        kp = self.context.keygen()
        self._keys = kp
        self._pk = kp.pk
        self._sk = kp.sk

    def relinKeyGen(self, bitCount: int = 1, size: int = 5):
        """Generate relinearization keys (reference intent at
        FLPyfhelin.py:357-364; its version is dead code with a NameError)."""
        if self._sk is None:
            raise ValueError("keyGen must run before relinKeyGen")
        if self._keys is None:  # keys restored from bytes, not keyGen()
            self._keys = KeyPair(sk=self._sk, pk=self._pk)
        self._keys.relin = self.context.relin_keygen(self._sk)

    # ----- scalar API (reference-exact shape) -----
    def encryptFrac(self, value: float) -> PyCtxt:
        """One scalar -> one ciphertext (reference FLPyfhelin.py:217).
        Kept for API parity; the batched tensor path is encrypt_tensor."""
        if self._pk is None:
            raise ValueError("no public key")
        pt = self.context.encode(np.array([float(value)]))
        return PyCtxt(self.context.encrypt(pt, self._pk), self)

    def decryptFrac(self, ct: PyCtxt) -> float:
        if self._sk is None:
            raise ValueError("no secret key")
        pt = self.context.decrypt(ct._ct, self._sk)
        return float(self.context.decode(pt, 1)[..., 0])

    # ----- batched tensor API (the hot path) -----
    def encrypt_tensor(self, vec: torch.Tensor) -> CtxtTensor:
        if self._pk is None:
            raise ValueError("no public key")
        return self.context.encrypt_tensor(vec, self._pk)

    def decrypt_tensor(self, ct: CtxtTensor) -> torch.Tensor:
        if self._sk is None:
            raise ValueError("no secret key")
        return self.context.decrypt_tensor(ct, self._sk)

    # ----- serialization (reference FLPyfhelin.py:337-338, 352-353, 257-259) -----
    def to_bytes_context(self) -> bytes:
        return _pack_header(b"CTX\x00", self.context.cfg)

    def from_bytes_context(self, buf: bytes):
        cfg, _ = _unpack_header(buf, b"CTX\x00")
        self.contextGen_cfg(cfg, device=self._device)

    def _key_bytes(self, kind: bytes, t: torch.Tensor) -> bytes:
        head = _pack_header(kind, self.context.cfg)
        body = io.BytesIO()
        np.save(body, t.cpu().numpy(), allow_pickle=False)
        return head + body.getvalue()

    def _key_from_bytes(self, kind: bytes, buf: bytes) -> torch.Tensor:
        cfg, off = _unpack_header(buf, kind)
        if self._ctx is None:
            self.contextGen_cfg(cfg, device=self._device)
        arr = np.load(io.BytesIO(buf[off:]), allow_pickle=False)
        return torch.from_numpy(arr).to(self.context.device)

    def to_bytes_publicKey(self) -> bytes:
        if self._pk is None:
            raise ValueError("no public key")
        return self._key_bytes(b"PK\x00\x00", self._pk)

    def from_bytes_publicKey(self, buf: bytes):
        self._pk = self._key_from_bytes(b"PK\x00\x00", buf)

    def to_bytes_secretKey(self) -> bytes:
        if self._sk is None:
            raise ValueError("no secret key")
        return self._key_bytes(b"SK\x00\x00", self._sk)

    def from_bytes_secretKey(self, buf: bytes):
        self._sk = self._key_from_bytes(b"SK\x00\x00", buf)

    # Pyfhel pickles whole objects in the reference's export dict
    # (FLPyfhelin.py:233); support that directly. The SECRET key is never
    # pickled: the reference's C-backed pickled Pyfhel carries no usable key
    # material, and a pickled object lands in 'public' artifacts
    # (publickey.pickle, per-client uploads) that the pk-only aggregation
    # server reads — sk must travel only via the explicit
    # to_bytes_secretKey/privatekey.pickle path (hefl/fl/keys.py).
    def __getstate__(self):
        state = {"cfg": None, "pk": None}
        if self._ctx is not None:
            state["cfg"] = self._ctx.cfg
        if self._pk is not None:
            state["pk"] = self._pk.cpu().numpy()
        return state

    def __setstate__(self, state):
        self.__init__()
        if state["cfg"] is not None:
            self.contextGen_cfg(state["cfg"])
        if state["pk"] is not None:
            self._pk = torch.from_numpy(state["pk"])
        if state.get("sk") is not None:  # legacy pickles only
            self._sk = torch.from_numpy(state["sk"])
