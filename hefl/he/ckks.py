"""RNS-CKKS scheme: context, keys, encrypt/decrypt, homomorphic ops.

From-scratch replacement for the reference's Pyfhel 2.3.1 -> SEAL 2.3 BFV
stack (FLPyfhelin.py:330-344 contextGen/keyGen; :217 encryptFrac; :295
decryptFrac; :381 ct+ct; :385 ct*plain). Differences by design (SURVEY.md
section 0): CKKS instead of BFV+FractionalEncoder, and slot packing — one
ciphertext carries n/2 weights instead of one scalar per ciphertext.

The scheme logic is written once, device-agnostic, over a small backend
interface (NTT + pointwise modular ops). CpuBackend = exact big-int numpy
(the test oracle); GpuBackend = hand-written HIP kernels on MI355X
(hefl/csrc/ntt.hip), ciphertexts resident in HBM as int64 tensors [.., 2, L, n].
Ciphertext data layout is all-reduce-ready: limb values < 2**60 so an int64
SUM over <= 8 clients cannot overflow (lazy reduction; hefl/fl/secure.py).
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import List, Optional, Sequence

import numpy as np
import torch

from ..config import HEConfig
from .encoder import Encoder
from .ntt_cpu import NttTables, fwd_ntt, inv_ntt
from .primes import gen_prime_chain

_TABLE_CACHE = {}


def _tables(q: int, n: int) -> NttTables:
    key = (q, n)
    if key not in _TABLE_CACHE:
        _TABLE_CACHE[key] = NttTables(q, n)
    return _TABLE_CACHE[key]


# ---------------------------------------------------------------------------
# Backends
# ---------------------------------------------------------------------------

class CpuBackend:
    """Exact reference backend (numpy object ints)."""

    def __init__(self, primes: Sequence[int], n: int):
        self.primes = list(primes)
        self.n = n
        self.tb = [_tables(q, n) for q in primes]

    def ntt(self, x: torch.Tensor, limb: int, inverse: bool = False) -> torch.Tensor:
        a = x.cpu().numpy().astype(object)
        tb = self.tb[limb]
        out = inv_ntt(a, tb) if inverse else fwd_ntt(a, tb)
        return torch.from_numpy(out.astype(np.int64))

    def modmul(self, a: torch.Tensor, b: torch.Tensor, limb: int) -> torch.Tensor:
        q = self.primes[limb]
        r = (a.cpu().numpy().astype(object) * b.cpu().numpy().astype(object)) % q
        return torch.from_numpy(r.astype(np.int64))

    def modmul_scalar(self, a: torch.Tensor, s: int, limb: int) -> torch.Tensor:
        q = self.primes[limb]
        r = (a.cpu().numpy().astype(object) * (s % q)) % q
        return torch.from_numpy(r.astype(np.int64))


class GpuBackend:
    """MI355X backend: batched HIP NTT + Barrett pointwise kernels (hefl._C)."""

    def __init__(self, primes: Sequence[int], n: int, device: torch.device):
        import hefl
        self._C = hefl.load_extension()
        self.primes = list(primes)
        self.n = n
        self.device = device
        # Device-side twiddle tables: [L, n] fwd / inv (+ Shoup companions),
        # built once from the exact CPU tables.
        w = torch.stack([torch.from_numpy(_tables(q, n).w.astype(np.int64))
                         for q in primes])
        winv = torch.stack([torch.from_numpy(_tables(q, n).winv.astype(np.int64))
                            for q in primes])
        qs = torch.tensor(self.primes, dtype=torch.int64)
        ninv = torch.tensor([_tables(q, n).n_inv for q in primes], dtype=torch.int64)

        def shoup(v, q):
            return torch.from_numpy(
                ((v.numpy().astype(object) << 64) // q).astype(np.uint64).astype(np.int64))

        self.w = w.to(device)
        self.w_shoup = torch.stack([shoup(w[i], q) for i, q in enumerate(primes)]).to(device)
        self.winv = winv.to(device)
        self.winv_shoup = torch.stack([shoup(winv[i], q) for i, q in enumerate(primes)]).to(device)
        self.qs = qs.to(device)
        self.ninv = ninv.to(device)
        self.ninv_shoup = torch.stack(
            [shoup(ninv[i:i + 1], q) for i, q in enumerate(primes)]).reshape(-1).to(device)
        # Barrett ratio words floor(2^128/q) per limb, for fused modmul
        words = []
        for q in primes:
            r = ((1 << 128) - 1) // q
            words.append([np.int64(np.uint64(r & ((1 << 64) - 1))),
                          np.int64(np.uint64(r >> 64))])
        self.ratio_words = torch.from_numpy(
            np.array(words, dtype=np.int64)).to(device)

    @staticmethod
    def _fresh(x: torch.Tensor) -> torch.Tensor:
        """Materialize a tensor the in-place NTT kernels may own: one copy,
        not two — .contiguous() already copies non-contiguous slices, so
        clone() only when it was a no-op (round-2 profile: the redundant
        clone showed up as ~2.5% of config #5 in hipMemcpy DtoD)."""
        c = x.contiguous()
        return c.clone() if c.data_ptr() == x.data_ptr() else c

    def ntt(self, x: torch.Tensor, limb: int, inverse: bool = False) -> torch.Tensor:
        out = self._fresh(x)
        flat = out.reshape(-1, self.n)
        if inverse:
            self._C.intt_batch(flat, self.winv[limb], self.winv_shoup[limb],
                               int(self.primes[limb]), int(self.ninv[limb]),
                               int(self.ninv_shoup[limb]))
        else:
            self._C.ntt_batch(flat, self.w[limb], self.w_shoup[limb],
                              int(self.primes[limb]))
        return out

    def modmul(self, a: torch.Tensor, b: torch.Tensor, limb: int) -> torch.Tensor:
        return self._C.modmul(a.contiguous(), b.contiguous(), int(self.primes[limb]))

    def modmul_scalar(self, a: torch.Tensor, s: int, limb: int) -> torch.Tensor:
        return self._C.modmul_scalar(a.contiguous(), int(s % self.primes[limb]),
                                     int(self.primes[limb]))

    # ----- fused multi-limb ops: one launch across the first L limbs -----
    def ntt_all(self, x: torch.Tensor, inverse: bool = False) -> torch.Tensor:
        """x [..., L, n] -> NTT per limb in ONE fused launch set."""
        L = x.shape[-2]
        out = self._fresh(x)
        flat = out.reshape(-1, L, self.n)
        if inverse:
            self._C.intt_limbs(flat, self.winv, self.winv_shoup, self.qs,
                               self.ninv, self.ninv_shoup, L)
        else:
            self._C.ntt_limbs(flat, self.w, self.w_shoup, self.qs, L)
        return out

    def modmul_limbs(self, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
        """a [..., L, n] x b ([..., L, n] or [L, n] tiling a), per-limb primes."""
        L = a.shape[-2]
        return self._C.modmul_limbs(a.contiguous(), b.contiguous(), self.qs,
                                    self.ratio_words, L, self.n)

    def modmul_scalar_limbs(self, a: torch.Tensor, scalars) -> torch.Tensor:
        """a [..., L, n] x per-limb plain scalars (python ints)."""
        L = a.shape[-2]
        sc, sh = [], []
        for i in range(L):
            q = self.primes[i]
            v = int(scalars[i]) % q
            sc.append(v)
            sh.append(np.int64(np.uint64((v << 64) // q)))
        dev = a.device
        sct = torch.tensor(sc, dtype=torch.int64, device=dev)
        sht = torch.tensor(np.array(sh, dtype=np.int64), device=dev)
        return self._C.modmul_scalar_limbs(a.contiguous(), sct, sht,
                                           self.qs, L, self.n)

    # ----- fused per-limb pointwise glue (replaces torch.remainder paths) ---
    def modadd_limbs(self, a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
        return self._C.modadd_limbs(a.contiguous(), b.contiguous(), self.qs,
                                    a.shape[-2], self.n)

    def modadd3_limbs(self, a, b, c) -> torch.Tensor:
        return self._C.modadd3_limbs(a.contiguous(), b.contiguous(),
                                     c.contiguous(), self.qs, a.shape[-2],
                                     self.n)

    def modsub_limbs(self, a: torch.Tensor, b: torch.Tensor,
                     L: Optional[int] = None) -> torch.Tensor:
        """out[.., l, :] = a[.., l, :] - b over the first L limbs (a may
        carry more limbs than L — the rescale slice without a copy)."""
        aL = a.shape[-2]
        L = aL if L is None else L
        return self._C.modsub_limbs(a.contiguous(), b.contiguous(), self.qs,
                                    L, aL, self.n)

    def bcast_center_mod(self, x: torch.Tensor, qc: int, L: int) -> torch.Tensor:
        """x [..., n] -> [..., L, n]: center mod qc (qc=0: already centered
        signed), then per-limb Barrett reduce. One launch replacing
        torch.where + broadcast remainder."""
        return self._C.bcast_center_mod(x.contiguous(), qc, self.qs,
                                        self.ratio_words, L)


# ---------------------------------------------------------------------------
# Data containers
# ---------------------------------------------------------------------------

@dataclass
class Plaintext:
    data: torch.Tensor  # int64 [..., L, n], NTT form
    scale: float


@dataclass
class Ciphertext:
    """One CKKS ciphertext: data int64 [2, L, n] (c0, c1), NTT form."""
    data: torch.Tensor
    scale: float

    @property
    def level(self) -> int:
        return self.data.shape[-2]

    def clone(self) -> "Ciphertext":
        return Ciphertext(self.data.clone(), self.scale)


@dataclass
class CtxtTensor:
    """Batched ciphertexts for one flat weight vector: int64 [B, 2, L, n]."""
    data: torch.Tensor
    scale: float
    count: int  # number of packed real values

    @property
    def level(self) -> int:
        return self.data.shape[-2]


@dataclass
class KeyPair:
    sk: torch.Tensor          # int64 [L, n] NTT form
    pk: torch.Tensor          # int64 [2, L, n] NTT form: (b, a)
    relin: Optional[torch.Tensor] = None  # [dnum, 2, L, n] later


# ---------------------------------------------------------------------------
# Context
# ---------------------------------------------------------------------------

class CKKSContext:
    def __init__(self, cfg: HEConfig, device: str = "cpu"):
        self.cfg = cfg
        self.n = cfg.m
        self.slots = cfg.m // 2
        # chain primes + ONE special prime for hybrid key-switching (the
        # relinearization path, configs #3/#5); special prime is never part
        # of a ciphertext level, only of relin keys.
        if not cfg.q_bits:
            raise ValueError("q_bits must name at least one chain prime "
                             "(L >= 1); an empty chain cannot encrypt")
        allp = gen_prime_chain(self.n, tuple(cfg.q_bits) + (60,))
        self.primes: List[int] = allp[:-1]
        self.special: int = allp[-1]
        self.all_primes: List[int] = allp
        self.L = len(self.primes)
        self.scale = float(2 ** cfg.scale_bits)
        self.encoder = Encoder(self.n)
        self.device = torch.device(device)
        if self.device.type == "cuda":
            self.backend = GpuBackend(self.all_primes, self.n, self.device)
        else:
            self.backend = CpuBackend(self.all_primes, self.n)
        self._cpu_rng = np.random.default_rng(cfg.seed)
        # security check against the HE standard (the reference passes
        # sec=128 to Pyfhel, FLPyfhelin.py:332; here it is verified):
        # total modulus = chain + special prime
        from .primes import max_logqp_128
        total_bits = sum(q.bit_length() for q in self.all_primes)
        limit = max_logqp_128(self.n)
        self.secure_128 = total_bits <= limit
        if cfg.sec >= 128 and not self.secure_128 and self.n >= 1024:
            import warnings
            warnings.warn(
                f"CKKS parameters (n={self.n}, logQP={total_bits}) exceed the "
                f"HE-standard 128-bit bound ({limit}); reduce q_bits or raise m",
                stacklevel=2)

    def reseed(self, salt: int) -> None:
        """Re-seed encryption randomness (e.g. per FL rank) after a shared-seed
        keygen, so ciphertext noise is independent across clients."""
        seed = None if self.cfg.seed is None else self.cfg.seed + 1000003 * (salt + 1)
        self._cpu_rng = np.random.default_rng(seed)
        if self.device.type == "cuda":
            g = torch.Generator(device=self.device)
            if seed is not None:
                g.manual_seed(seed)
            self._gpu_gen = g

    @property
    def gpu_gen(self):
        if getattr(self, "_gpu_gen", None) is None:
            g = torch.Generator(device=self.device)
            if self.cfg.seed is not None:
                g.manual_seed(self.cfg.seed)
            self._gpu_gen = g
        return self._gpu_gen

    # ----- helpers -----
    def _q(self, limb: int) -> int:
        return self.all_primes[limb]  # index L = the special prime

    def _modadd(self, a: torch.Tensor, b: torch.Tensor, limbs: Sequence[int]) -> torch.Tensor:
        if self.device.type == "cuda" and list(limbs) == list(range(len(limbs))):
            # fused per-limb HIP kernel (inputs are valid residues < q)
            return self.backend.modadd_limbs(a, b)
        qs = torch.tensor([self._q(i) for i in limbs], dtype=torch.int64,
                          device=a.device)
        shape = [1] * a.dim()
        shape[-2] = len(limbs)
        return torch.remainder(a + b, qs.view(shape))

    def _modsub(self, a: torch.Tensor, b: torch.Tensor, limbs: Sequence[int]) -> torch.Tensor:
        if self.device.type == "cuda" and list(limbs) == list(range(len(limbs))):
            return self.backend.modsub_limbs(a, b)
        qs = torch.tensor([self._q(i) for i in limbs], dtype=torch.int64,
                          device=a.device)
        shape = [1] * a.dim()
        shape[-2] = len(limbs)
        return torch.remainder(a - b, qs.view(shape))

    # ----- sampling (coefficient domain) -----
    # Keygen always samples on the host RNG (deterministic across CPU/GPU
    # contexts with the same seed — all FL ranks must derive the SAME keys).
    # Per-message encryption noise samples on the DEVICE RNG when on GPU:
    # host sampling of u/e0/e1 for a packed weight tensor costs tens of ms
    # of numpy + transfer, device sampling is microseconds.
    def _sample_ternary(self, shape, host: bool = False) -> torch.Tensor:
        # {-1, 0, 1} uniform (hamming-weight variant not needed at these n)
        if not host and self.device.type == "cuda":
            return torch.randint(-1, 2, tuple(shape), generator=self.gpu_gen,
                                 device=self.device, dtype=torch.int64)
        v = torch.from_numpy(self._cpu_rng.integers(-1, 2, size=shape))
        return v.to(torch.int64)

    def _sample_err(self, shape, eta: int = 21, host: bool = False) -> torch.Tensor:
        # centered binomial, sigma = sqrt(eta/2) ~= 3.24 (SEAL sigma 3.2)
        if not host and self.device.type == "cuda":
            # one uniform 64-bit draw per coefficient; the HIP kernel takes
            # popcount(bits[0:21]) - popcount(bits[21:42])
            bits = torch.randint(-(2 ** 63), 2 ** 63 - 1, tuple(shape),
                                 generator=self.gpu_gen, device=self.device,
                                 dtype=torch.int64)
            return self.backend._C.cbd21(bits)
        b = self._cpu_rng.integers(0, 2, size=(eta,) + tuple(shape)).sum(axis=0)
        b2 = self._cpu_rng.integers(0, 2, size=(eta,) + tuple(shape)).sum(axis=0)
        return torch.from_numpy(b - b2).to(torch.int64)

    def _sample_uniform(self, shape_limbs) -> torch.Tensor:
        """Uniform in [0, q_i) per limb, NTT domain. shape_limbs: [..., L, n]."""
        outs = []
        for i in range(shape_limbs[-2]):
            hi = self._q(i)
            outs.append(torch.from_numpy(
                self._cpu_rng.integers(0, hi, size=shape_limbs[:-2] + (shape_limbs[-1],),
                                       dtype=np.int64)))
        return torch.stack(outs, dim=-2)

    def _to_rns_ntt(self, coeffs: torch.Tensor, nlimbs: Optional[int] = None) -> torch.Tensor:
        """Small centered int64 coeffs [..., n] -> NTT-form RNS [..., L, n]."""
        nlimbs = self.L if nlimbs is None else nlimbs
        if self.device.type == "cuda":
            rem = self.backend.bcast_center_mod(coeffs.to(self.device), 0,
                                                nlimbs)
            return self.backend.ntt_all(rem)
        out = []
        for i in range(nlimbs):
            q = self._q(i)
            ci = torch.remainder(coeffs, q).to(self.device)
            out.append(self.backend.ntt(ci, i))
        return torch.stack(out, dim=-2)

    # ----- keys -----
    def keygen(self) -> KeyPair:
        n = self.n
        s = self._sample_ternary((n,), host=True)
        e = self._sample_err((n,), host=True)
        sk = self._to_rns_ntt(s)                       # [L, n]
        ehat = self._to_rns_ntt(e)
        a = self._sample_uniform((self.L, n)).to(self.device)   # NTT-domain uniform
        b = torch.empty_like(a)
        for i in range(self.L):
            as_ = self.backend.modmul(a[i], sk[i], i)
            b[i] = torch.remainder(-(as_ + ehat[i]), self._q(i))
        pk = torch.stack([b, a])                       # [2, L, n]
        return KeyPair(sk=sk, pk=pk)

    # ----- encode / decode -----
    def encode(self, vals: np.ndarray, nlimbs: Optional[int] = None,
               scale: Optional[float] = None) -> Plaintext:
        """vals: real [..., k<=slots] -> NTT-form RNS plaintext."""
        scale = self.scale if scale is None else scale
        nlimbs = self.L if nlimbs is None else nlimbs
        if self.device.type == "cuda":
            # full-device path: special FFT + rounding on the GPU, then one
            # fused multi-limb remainder + NTT
            vt = (vals if torch.is_tensor(vals)
                  else torch.from_numpy(np.asarray(vals, dtype=np.float64)))
            ct = self.encoder.encode_torch(vt.to(self.device), scale)
            return Plaintext(self._to_rns_ntt(ct, nlimbs), scale)
        coeffs = self.encoder.encode(np.asarray(vals, dtype=np.float64), scale)
        out = []
        if coeffs.dtype == np.int64:
            # fast path: one upload, per-limb remainder + NTT on device
            ct = torch.from_numpy(coeffs).to(self.device)
            for i in range(nlimbs):
                out.append(self.backend.ntt(torch.remainder(ct, self._q(i)), i))
        else:
            for i in range(nlimbs):
                q = self._q(i)
                ci = np.mod(coeffs, q)  # object -> [0, q)
                ci_t = torch.from_numpy(ci.astype(np.int64)).to(self.device)
                out.append(self.backend.ntt(ci_t, i))
        return Plaintext(torch.stack(out, dim=-2), scale)

    def decode(self, pt: Plaintext, k: int) -> np.ndarray:
        data = pt.data
        nlimbs = data.shape[-2]
        if nlimbs == 1:
            # fast path (the FedAvg pipeline decrypts at level 1): center on
            # device, f64 is exact for plaintext magnitudes << 2^53
            q = self._q(0)
            c = self.backend.ntt(data[..., 0, :], 0, inverse=True)
            cent = torch.where(c > q // 2, c - q, c)
            if self.device.type == "cuda":
                return self.encoder.decode_torch(cent, pt.scale, k)
            return self.encoder.decode(cent.cpu().numpy(), pt.scale, k)
        # multi-limb fast path: when the plaintext magnitude is << q0/2 (true
        # for any decode of real-valued weights at scale <= 2^45), the
        # centered limb-0 residue IS the integer value — checked against
        # limb 1 on a sample, falling back to exact big-int CRT on mismatch.
        q0 = self._q(0)
        c0 = self.backend.ntt(data[..., 0, :], 0, inverse=True)
        cent0 = torch.where(c0 > q0 // 2, c0 - q0, c0)
        if cent0.abs().max().item() < q0 // 4:
            q1 = self._q(1)
            c1 = self.backend.ntt(data[..., 1, :], 1, inverse=True)
            cent1 = torch.where(c1 > q1 // 2, c1 - q1, c1)
            if torch.equal(torch.remainder(cent0, q1), torch.remainder(cent1, q1)):
                if self.device.type == "cuda":
                    return self.encoder.decode_torch(cent0, pt.scale, k)
                return self.encoder.decode(cent0.cpu().numpy(), pt.scale, k)
        # exact big-int CRT path (plaintext too large for limb-0 shortcut)
        coeff_limbs = []
        for i in range(nlimbs):
            c = self.backend.ntt(data[..., i, :], i, inverse=True)
            coeff_limbs.append(c.cpu().numpy().astype(object))
        qs = [self._q(i) for i in range(nlimbs)]
        Q = math.prod(qs)
        x = 0
        for i, q in enumerate(qs):
            Qi = Q // q
            hi = pow(Qi % q, -1, q)
            x = x + coeff_limbs[i] * ((Qi * hi) % Q)
        x = np.mod(x, Q)
        centered = np.where(x > Q // 2, x - Q, x)
        return self.encoder.decode(centered, pt.scale, k)

    # ----- encrypt / decrypt (single ct or batched [..., 2, L, n]) -----
    def encrypt(self, pt: Plaintext, pk: torch.Tensor) -> Ciphertext:
        data = self._encrypt_data(pt.data, pk)
        return Ciphertext(data, pt.scale)

    def _encrypt_data(self, ptdata: torch.Tensor, pk: torch.Tensor) -> torch.Tensor:
        lead = ptdata.shape[:-2]
        nlimbs = ptdata.shape[-2]
        n = self.n
        u = self._to_rns_ntt(self._sample_ternary(lead + (n,)), nlimbs)
        e0 = self._to_rns_ntt(self._sample_err(lead + (n,)), nlimbs)
        e1 = self._to_rns_ntt(self._sample_err(lead + (n,)), nlimbs)
        if self.device.type == "cuda":
            bu = self.backend.modmul_limbs(u, pk[0, :nlimbs])
            au = self.backend.modmul_limbs(u, pk[1, :nlimbs])
            c0 = self.backend.modadd3_limbs(bu, e0, ptdata.contiguous())
            c1 = self.backend.modadd_limbs(au, e1)
            return torch.stack([c0, c1], dim=-3)
        c0 = torch.empty_like(ptdata)
        c1 = torch.empty_like(ptdata)
        for i in range(nlimbs):
            q = self._q(i)
            bu = self.backend.modmul(pk[0, i].expand(lead + (n,)).contiguous(),
                                     u[..., i, :], i)
            au = self.backend.modmul(pk[1, i].expand(lead + (n,)).contiguous(),
                                     u[..., i, :], i)
            c0[..., i, :] = torch.remainder(bu + e0[..., i, :] + ptdata[..., i, :], q)
            c1[..., i, :] = torch.remainder(au + e1[..., i, :], q)
        return torch.stack([c0, c1], dim=-3)  # [..., 2, L, n]

    def decrypt(self, ct: Ciphertext, sk: torch.Tensor) -> Plaintext:
        return Plaintext(self._decrypt_data(ct.data, sk), ct.scale)

    def _decrypt_data(self, ctdata: torch.Tensor, sk: torch.Tensor) -> torch.Tensor:
        nlimbs = ctdata.shape[-2]
        c0 = ctdata[..., 0, :, :]
        c1 = ctdata[..., 1, :, :]
        if self.device.type == "cuda":
            cs = self.backend.modmul_limbs(c1.contiguous(), sk[:nlimbs])
            return self.backend.modadd_limbs(c0.contiguous(), cs)
        out = torch.empty_like(c0)
        for i in range(nlimbs):
            cs = self.backend.modmul(c1[..., i, :], sk[i].expand_as(c1[..., i, :]).contiguous(), i)
            out[..., i, :] = torch.remainder(c0[..., i, :] + cs, self._q(i))
        return out

    # ----- homomorphic ops -----
    @staticmethod
    def _check_add_scales(sa: float, sb: float) -> None:
        # adding ciphertexts at different scales silently decodes to garbage
        # (the slots are Delta-scaled integers; sums only make sense at one
        # Delta) — fail loudly instead
        if abs(sa - sb) > 1e-9 * max(sa, sb):
            raise ValueError(
                f"ciphertext scales differ ({sa} vs {sb}); rescale or "
                f"re-encode one operand before adding")

    def add(self, a: Ciphertext, b: Ciphertext) -> Ciphertext:
        assert a.level == b.level
        self._check_add_scales(a.scale, b.scale)
        limbs = list(range(a.level))
        return Ciphertext(self._modadd(a.data, b.data, limbs), max(a.scale, b.scale))

    def mul_scalar_data(self, data: torch.Tensor, x: float,
                        scale: Optional[float] = None) -> torch.Tensor:
        """Multiply ct data [..., 2, L, n] by encoded scalar round(x * Delta)."""
        delta = self.scale if scale is None else scale
        sc = int(round(x * delta))
        nlimbs = data.shape[-2]
        if self.device.type == "cuda":
            return self.backend.modmul_scalar_limbs(
                data.contiguous(), [sc] * nlimbs)
        out = torch.empty_like(data)
        for i in range(nlimbs):
            q = self._q(i)
            out[..., i, :] = self.backend.modmul_scalar(data[..., i, :], sc % q, i)
        return out

    def mul_scalar(self, ct: Ciphertext, x: float) -> Ciphertext:
        return Ciphertext(self.mul_scalar_data(ct.data, x), ct.scale * self.scale)

    def mul_plain(self, ct: Ciphertext, pt: Plaintext) -> Ciphertext:
        nlimbs = ct.level
        out = torch.empty_like(ct.data)
        for i in range(nlimbs):
            for c in range(2):
                out[..., c, i, :] = self.backend.modmul(
                    ct.data[..., c, i, :], pt.data[..., i, :], i)
        return Ciphertext(out, ct.scale * pt.scale)

    def rescale_data(self, data: torch.Tensor) -> torch.Tensor:
        """Drop the last limb: exact RNS division by q_last with rounding.
        data: [..., 2, L, n] -> [..., 2, L-1, n]."""
        nlimbs = data.shape[-2]
        assert nlimbs >= 2, "cannot rescale at level 1"
        last = nlimbs - 1
        qL = self._q(last)
        qL_half = qL // 2
        if self.device.type == "cuda":
            # fully fused: INTT of the last limb, one center+broadcast-reduce
            # launch, one fused NTT set, one strided-slice modsub (no
            # materialized slice), one scalar-limbs multiply — zero
            # at::native kernels on this path
            cl = self.backend.ntt(data[..., last, :], last,
                                  inverse=True)
            r = self.backend.bcast_center_mod(cl, qL, last)
            r_ntt = self.backend.ntt_all(r)
            diff = self.backend.modsub_limbs(data.contiguous(), r_ntt, last)
            inv = [pow(qL % self._q(i), -1, self._q(i)) for i in range(last)]
            return self.backend.modmul_scalar_limbs(diff, inv)
        # coefficient-domain last limb, centered for round-to-nearest
        cl = self.backend.ntt(data[..., last, :], last,
                              inverse=True)
        cl_c = torch.where(cl > qL_half, cl - qL, cl)
        out = data[..., :last, :].clone()
        for i in range(last):
            q = self._q(i)
            r = torch.remainder(cl_c, q)
            r_ntt = self.backend.ntt(r, i)
            diff = torch.remainder(out[..., i, :] - r_ntt, q)
            inv_qL = pow(qL % q, -1, q)
            out[..., i, :] = self.backend.modmul_scalar(diff, inv_qL, i)
        return out

    def rescale(self, ct: Ciphertext) -> Ciphertext:
        new = self.rescale_data(ct.data)
        return Ciphertext(new, ct.scale / float(self._q(ct.level - 1)))

    # ----- batched tensor API (flat weight vectors) -----
    def encrypt_tensor(self, vec: torch.Tensor, pk: torch.Tensor) -> CtxtTensor:
        """Slot-pack a flat fp32 vector into ceil(len/slots) ciphertexts."""
        count = vec.numel()
        B = (count + self.slots - 1) // self.slots
        if self.device.type == "cuda":
            buf = torch.zeros(B * self.slots, dtype=torch.float64,
                              device=self.device)
            buf[:count] = vec.detach().double().reshape(-1).to(self.device)
            pt = self.encode(buf.reshape(B, self.slots))
        else:
            buf = np.zeros((B, self.slots), dtype=np.float64)
            buf.reshape(-1)[:count] = vec.detach().float().cpu().numpy().reshape(-1)
            pt = self.encode(buf)                 # [B, L, n]
        data = self._encrypt_data(pt.data, pk)    # [B, 2, L, n]
        return CtxtTensor(data, pt.scale, count)

    def decrypt_tensor(self, ct: CtxtTensor, sk: torch.Tensor) -> torch.Tensor:
        pt = self._decrypt_data(ct.data, sk)      # [B, L, n]
        vals = self.decode(Plaintext(pt, ct.scale), self.slots)  # [B, slots]
        if torch.is_tensor(vals):
            return vals.reshape(-1)[:ct.count].to(torch.float32)
        flat = torch.from_numpy(np.ascontiguousarray(vals.reshape(-1)[:ct.count]))
        return flat.to(torch.float32)

    def add_tensor(self, a: CtxtTensor, b: CtxtTensor) -> CtxtTensor:
        assert a.count == b.count and a.level == b.level
        self._check_add_scales(a.scale, b.scale)
        limbs = list(range(a.level))
        return CtxtTensor(self._modadd(a.data, b.data, limbs),
                          max(a.scale, b.scale), a.count)

    def mul_scalar_tensor(self, ct: CtxtTensor, x: float) -> CtxtTensor:
        return CtxtTensor(self.mul_scalar_data(ct.data, x),
                          ct.scale * self.scale, ct.count)

    def rescale_tensor(self, ct: CtxtTensor) -> CtxtTensor:
        return CtxtTensor(self.rescale_data(ct.data),
                          ct.scale / float(self._q(ct.level - 1)), ct.count)

    # ----- relinearization (ct x ct support; reference intent at
    # FLPyfhelin.py:357-364 gen_rekey — dead code there, real here) -----
    def _sk_extended(self, sk: torch.Tensor) -> torch.Tensor:
        """Recover the ternary secret from sk's limb 0 and re-embed over
        all L+1 limbs (chain + special) for key generation."""
        q0 = self._q(0)
        s = self.backend.ntt(sk[0], 0, inverse=True).cpu()
        s = torch.where(s > q0 // 2, s - q0, s)  # ternary {-1,0,1}
        return self._to_rns_ntt(s, self.L + 1)

    def relin_keygen(self, sk: torch.Tensor) -> torch.Tensor:
        """Hybrid key-switching keys (digits = RNS limbs, one special prime P).

        rlk[d] = (-(a_d*s + e_d) + P*q̃_d*s², a_d) over the L+1 limbs
        {q_0..q_{L-1}, P}. In RNS, P*q̃_d*s² is (P mod q_d)*s² at limb d and
        0 elsewhere (q̃_d ≡ δ_{d,i} mod q_i, P ≡ 0 mod P). Key-switch noise
        shrinks by 1/P at the final mod-down, which is what makes per-limb
        digits sound (classic per-limb BV without P has q_max-sized noise).
        Returns int64 [L, 2, L+1, n] in NTT form.
        """
        L, n = self.L, self.n
        Lp = L + 1
        P = self.special
        sk_ext = self._sk_extended(sk)            # [L+1, n]
        s2 = torch.empty_like(sk_ext)
        for i in range(Lp):
            s2[i] = self.backend.modmul(sk_ext[i], sk_ext[i], i)
        rlk = torch.empty((L, 2, Lp, n), dtype=torch.int64, device=self.device)
        for d in range(L):
            a = self._sample_uniform((Lp, n)).to(self.device)
            e = self._to_rns_ntt(self._sample_err((n,), host=True), Lp)
            for i in range(Lp):
                q = self._q(i)
                b = torch.remainder(-(self.backend.modmul(a[i], sk_ext[i], i)
                                      + e[i]), q)
                if i == d:
                    b = torch.remainder(
                        b + self.backend.modmul_scalar(s2[i], P % q, i), q)
                rlk[d, 0, i] = b
                rlk[d, 1, i] = a[i]
        return rlk

    def _keyswitch(self, d2: torch.Tensor, rlk: torch.Tensor):
        """Key-switch NTT-form d2 [..., L, n] through rlk; returns the pair
        (ks0, ks1) each [..., L, n] to add to (c0, c1).

        GPU path is fully fused: one INTT set over all
        digits, one lift launch, one NTT set, ONE inner-product kernel over
        all digits x limbs, then a fused mod-down — ~12 launches total
        instead of the O(L*(L+1)) per-digit/per-limb Python loop."""
        L = d2.shape[-2]
        Lp = L + 1
        n = self.n
        if self.device.type == "cuda":
            be = self.backend
            c = be.ntt_all(d2, inverse=True)           # digits -> coeff domain
            dig = be.bcast_center_mod(c, 0, Lp)        # [..., L, Lp, n]
            dig = be.ntt_all(dig)                      # NTT over the Lp limbs
            acc0, acc1 = be._C.ks_inner(dig.contiguous(), rlk.contiguous(),
                                        be.qs, be.ratio_words, L, Lp, n)
            P = self.special
            inv = [pow(P % self._q(i), -1, self._q(i)) for i in range(L)]
            outs = []
            for acc in (acc0, acc1):
                cl = be.ntt(acc[..., L, :], L, inverse=True)
                r_ntt = be.ntt_all(be.bcast_center_mod(cl, P, L))
                diff = be.modsub_limbs(acc, r_ntt, L)
                outs.append(be.modmul_scalar_limbs(diff, inv))
            return outs[0], outs[1]
        lead = d2.shape[:-2]
        acc0 = torch.zeros(lead + (Lp, n), dtype=torch.int64, device=d2.device)
        acc1 = torch.zeros_like(acc0)
        for d in range(L):
            # digit d: coefficient-domain residues mod q_d, lifted to every limb
            c = self.backend.ntt(d2[..., d, :], d, inverse=True)
            for i in range(Lp):
                q = self._q(i)
                dig = self.backend.ntt(torch.remainder(c, q), i)
                acc0[..., i, :] = torch.remainder(
                    acc0[..., i, :] + self.backend.modmul(
                        dig, rlk[d, 0, i].expand_as(dig).contiguous(), i), q)
                acc1[..., i, :] = torch.remainder(
                    acc1[..., i, :] + self.backend.modmul(
                        dig, rlk[d, 1, i].expand_as(dig).contiguous(), i), q)
        # mod-down by the special prime P (exact division with rounding,
        # same structure as rescale_data but dropping the special limb)
        P = self.special
        P_half = P // 2
        outs = []
        for acc in (acc0, acc1):
            cl = self.backend.ntt(acc[..., L, :], L, inverse=True)
            cl_c = torch.where(cl > P_half, cl - P, cl)
            out = acc[..., :L, :].clone()
            for i in range(L):
                q = self._q(i)
                r_ntt = self.backend.ntt(torch.remainder(cl_c, q), i)
                diff = torch.remainder(out[..., i, :] - r_ntt, q)
                out[..., i, :] = self.backend.modmul_scalar(
                    diff, pow(P % q, -1, q), i)
            outs.append(out)
        return outs[0], outs[1]

    def mul_ct(self, a: Ciphertext, b: Ciphertext, rlk: torch.Tensor) -> Ciphertext:
        """ct x ct multiply + relinearize (no rescale; call rescale after)."""
        if rlk is None:
            raise ValueError(
                "ct x ct multiply needs relinearization keys — run "
                "relin_keygen(sk) (Pyfhel API: relinKeyGen) first")
        assert a.level == b.level
        L = a.level
        if self.device.type == "cuda":
            # one fused tensor-product launch (d0, d1, d2 together)
            d0, d1, d2 = self.backend._C.ct_mul(
                a.data.contiguous(), b.data.contiguous(), self.backend.qs,
                self.backend.ratio_words, L, self.n)
        else:
            a0, a1 = a.data[..., 0, :, :], a.data[..., 1, :, :]
            b0, b1 = b.data[..., 0, :, :], b.data[..., 1, :, :]
            d0 = torch.empty_like(a0)
            d1 = torch.empty_like(a0)
            d2 = torch.empty_like(a0)
            for i in range(L):
                q = self._q(i)
                d0[..., i, :] = self.backend.modmul(a0[..., i, :], b0[..., i, :], i)
                cross = (self.backend.modmul(a0[..., i, :], b1[..., i, :], i)
                         + self.backend.modmul(a1[..., i, :], b0[..., i, :], i))
                d1[..., i, :] = torch.remainder(cross, q)
                d2[..., i, :] = self.backend.modmul(a1[..., i, :], b1[..., i, :], i)
        ks0, ks1 = self._keyswitch(d2, rlk)
        limbs = list(range(L))
        c0 = self._modadd(d0, ks0, limbs)
        c1 = self._modadd(d1, ks1, limbs)
        return Ciphertext(torch.stack([c0, c1], dim=-3), a.scale * b.scale)

    def modreduce_tensor_(self, ct: CtxtTensor) -> CtxtTensor:
        """Reduce lazily-summed (int64) limb values back to [0, q_i) in place
        — the step after the RCCL all-reduce of raw coefficient tensors."""
        if self.device.type == "cuda":
            # branchless conditional-subtract kernel (sum of <= 8 residues)
            self.backend._C.modreduce_(
                ct.data, self.backend.qs[:ct.level].contiguous())
            return ct
        qs = torch.tensor(self.primes[:ct.level], dtype=torch.int64,
                          device=ct.data.device)
        ct.data.remainder_(qs.view(1, 1, -1, 1))
        return ct
