"""CPU reference negacyclic NTT over Z_q[X]/(X^n + 1).

Exact big-integer arithmetic via numpy object arrays — this is the numerics
oracle the HIP NTT kernels are validated against (tests/test_gpu_he.py), and
the CPU execution path of the HE layer (CPU plumbing config / tests).

Cooley-Tukey forward / Gentleman-Sande inverse with merged psi powers in
bit-reversed order (the scheme SEAL uses inside every encrypt/mult that the
reference invokes via Pyfhel, FLPyfhelin.py:217,295,381,385).
"""
from __future__ import annotations

import numpy as np

from .primes import bit_reverse, primitive_root_2n


class NttTables:
    def __init__(self, q: int, n: int):
        self.q = q
        self.n = n
        self.logn = n.bit_length() - 1
        psi = primitive_root_2n(q, n)
        self.psi = psi
        self.n_inv = pow(n, -1, q)
        psi_pows = [1] * n
        for i in range(1, n):
            psi_pows[i] = psi_pows[i - 1] * psi % q
        ipsi = pow(psi, -1, q)
        ipsi_pows = [1] * n
        for i in range(1, n):
            ipsi_pows[i] = ipsi_pows[i - 1] * ipsi % q
        # bit-reversed twiddle tables (index m+i at stage with m groups)
        self.w = np.array([psi_pows[bit_reverse(i, self.logn)] for i in range(n)],
                          dtype=object)
        self.winv = np.array([ipsi_pows[bit_reverse(i, self.logn)] for i in range(n)],
                             dtype=object)


def fwd_ntt(a: np.ndarray, tb: NttTables) -> np.ndarray:
    """Batched forward negacyclic NTT. a: object array [..., n] in [0,q)."""
    q, n = tb.q, tb.n
    shape = a.shape
    B = int(np.prod(shape[:-1], dtype=np.int64)) if len(shape) > 1 else 1
    a = a.reshape(B, n).copy()
    t = n
    m = 1
    while m < n:
        t //= 2
        s = tb.w[m:2 * m]  # [m]
        v = a.reshape(B, m, 2, t)
        V = (v[:, :, 1, :] * s[None, :, None]) % q
        U = v[:, :, 0, :].copy()
        v[:, :, 0, :] = (U + V) % q
        v[:, :, 1, :] = (U - V) % q
        m *= 2
    return a.reshape(shape)


def inv_ntt(a: np.ndarray, tb: NttTables) -> np.ndarray:
    """Batched inverse negacyclic NTT (inverse of fwd_ntt)."""
    q, n = tb.q, tb.n
    shape = a.shape
    B = int(np.prod(shape[:-1], dtype=np.int64)) if len(shape) > 1 else 1
    a = a.reshape(B, n).copy()
    t = 1
    m = n
    while m > 1:
        h = m // 2
        s = tb.winv[h:m]  # [h]
        v = a.reshape(B, h, 2, t)
        U = v[:, :, 0, :].copy()
        V = v[:, :, 1, :].copy()
        v[:, :, 0, :] = (U + V) % q
        v[:, :, 1, :] = ((U - V) * s[None, :, None]) % q
        t *= 2
        m = h
    a = (a * tb.n_inv) % q
    return a.reshape(shape)


def negacyclic_mul_naive(a, b, q: int, n: int):
    """O(n^2) schoolbook negacyclic product — the oracle's oracle (tests only)."""
    res = [0] * n
    for i, ai in enumerate(a):
        if ai == 0:
            continue
        for j, bj in enumerate(b):
            k = i + j
            if k < n:
                res[k] = (res[k] + ai * bj) % q
            else:
                res[k - n] = (res[k - n] - ai * bj) % q
    return np.array(res, dtype=object)
