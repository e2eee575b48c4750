"""Prime-chain and root-of-unity generation for RNS-CKKS.

Replaces SEAL's parameter selection behind Pyfhel's contextGen
(reference FLPyfhelin.py:332). All primes are NTT-friendly (q ≡ 1 mod 2n)
and < 2**60 so a lazy int64 sum over <= 8 clients never overflows during the
RCCL all-reduce (collectives row).
"""
from __future__ import annotations

from typing import List

_MR_BASES = (2, 3, 5, 7, 11, 13, 17, 19, 23, 29, 31, 37)  # deterministic < 3.3e24


def is_prime(n: int) -> bool:
    if n < 2:
        return False
    for p in _MR_BASES:
        if n % p == 0:
            return n == p
    d, r = n - 1, 0
    while d % 2 == 0:
        d //= 2
        r += 1
    for a in _MR_BASES:
        x = pow(a, d, n)
        if x in (1, n - 1):
            continue
        for _ in range(r - 1):
            x = x * x % n
            if x == n - 1:
                break
        else:
            return False
    return True


def gen_prime_chain(n: int, bit_sizes) -> List[int]:
    """Distinct primes q_i ≡ 1 (mod 2n), q_i just below 2**bits (q_i < 2**60)."""
    out: List[int] = []
    for bits in bit_sizes:
        if bits > 60:
            raise ValueError("limb primes must be < 2**60 (lazy all-reduce bound)")
        cand = (1 << bits) - ((1 << bits) - 1) % (2 * n) - 1  # largest ≡1 mod 2n below 2^bits
        cand += 1
        assert cand % (2 * n) == 1
        while cand > 1:
            if cand not in out and is_prime(cand):
                out.append(cand)
                break
            cand -= 2 * n
        else:
            raise RuntimeError(f"no prime found for {bits} bits, 2n={2*n}")
    return out


def primitive_root_2n(q: int, n: int) -> int:
    """psi with order exactly 2n mod q (psi**n == -1)."""
    assert (q - 1) % (2 * n) == 0
    e = (q - 1) // (2 * n)
    g = 2
    while True:
        psi = pow(g, e, q)
        if psi != 1 and pow(psi, n, q) == q - 1:
            return psi
        g += 1
        if g > 1000:
            raise RuntimeError("no primitive 2n-th root found")


def bit_reverse(x: int, bits: int) -> int:
    r = 0
    for _ in range(bits):
        r = (r << 1) | (x & 1)
        x >>= 1
    return r


# HE-standard (homomorphicencryption.org) maximum log2(Q*P) for classical
# 128-bit security with ternary secrets, per ring dimension.
HE_STD_128_CLASSICAL = {
    1024: 27, 2048: 54, 4096: 109, 8192: 218, 16384: 438, 32768: 881,
}


def max_logqp_128(n: int) -> int:
    """Largest total modulus (chain + special prime) bit count meeting
    classical 128-bit security at ring dimension n. Dimensions below 1024
    (test-only) return 0 — never secure."""
    return HE_STD_128_CLASSICAL.get(n, 0)
