"""CKKS canonical-embedding encoder (CPU reference, float64).

encode: C^{n/2} slot vector -> integer polynomial coefficients (scaled by
Delta) via the inverse special FFT; decode is the forward special FFT.
Structure follows the standard CKKS embedding with the 5^j rotation group.

The reference encodes ONE scalar per ciphertext (BFV FractionalEncoder,
FLPyfhelin.py:217); here a ciphertext packs n/2 real weights per ct — the
main algorithmic headroom over the reference.
"""
from __future__ import annotations

import numpy as np

from .primes import bit_reverse


class Encoder:
    def __init__(self, n: int):
        self.n = n
        self.slots = n // 2
        M = 2 * n  # 4 * slots
        self.M = M
        self.ksi = np.exp(2j * np.pi * np.arange(M) / M)  # ksi^k = e^{2πik/M}
        self.rot = [pow(5, j, M) for j in range(self.slots)]
        self.log_slots = self.slots.bit_length() - 1

    # --- special FFT pair (HEAAN-style butterflies over the 5^j group) ---
    def _bitrev(self, v: np.ndarray) -> np.ndarray:
        s = v.shape[-1]
        bits = s.bit_length() - 1
        perm = np.array([bit_reverse(i, bits) for i in range(s)])
        return v[..., perm]

    def fft_special(self, v: np.ndarray) -> np.ndarray:
        """Decode direction: coeff-side values -> slot values. v: [..., slots]."""
        v = self._bitrev(np.asarray(v, dtype=np.complex128))
        size = self.slots
        length = 2
        while length <= size:
            lenh, lenq = length // 2, length * 4
            idx = np.array([(self.rot[j] % lenq) * (self.M // lenq)
                            for j in range(lenh)])
            w = self.ksi[idx]  # [lenh]
            v = v.reshape(*v.shape[:-1], size // length, 2, lenh)
            u = v[..., 0, :]
            t = v[..., 1, :] * w
            v = np.stack([u + t, u - t], axis=-2)
            v = v.reshape(*v.shape[:-3], size)
            length *= 2
        return v

    def fft_special_inv(self, v: np.ndarray) -> np.ndarray:
        """Encode direction: slot values -> coeff-side values. v: [..., slots]."""
        v = np.asarray(v, dtype=np.complex128).copy()
        size = self.slots
        length = size
        while length >= 2:
            lenh, lenq = length // 2, length * 4
            idx = np.array([(lenq - (self.rot[j] % lenq)) * (self.M // lenq)
                            for j in range(lenh)])
            w = self.ksi[idx]
            v = v.reshape(*v.shape[:-1], size // length, 2, lenh)
            u = v[..., 0, :] + v[..., 1, :]
            t = (v[..., 0, :] - v[..., 1, :]) * w
            v = np.stack([u, t], axis=-2)
            v = v.reshape(*v.shape[:-3], size)
            length //= 2
        v = self._bitrev(v)
        return v / size

    # --- encode / decode ---
    def encode(self, vals: np.ndarray, scale: float) -> np.ndarray:
        """vals: real [..., k<=slots] -> integer coeffs [..., n] (Python ints,
        centered: may be negative)."""
        vals = np.asarray(vals, dtype=np.float64)
        pad = self.slots - vals.shape[-1]
        if pad < 0:
            raise ValueError("too many values for slot count")
        if pad:
            vals = np.concatenate(
                [vals, np.zeros(vals.shape[:-1] + (pad,))], axis=-1)
        z = self.fft_special_inv(vals)
        re = np.round(z.real * scale)
        im = np.round(z.imag * scale)
        peak = max(np.abs(re).max(initial=0.0), np.abs(im).max(initial=0.0))
        if peak < 2.0 ** 52:
            # int64 fast path (exact: f64 integers below 2^53)
            return np.concatenate([re, im], axis=-1).astype(np.int64)
        coeffs = np.concatenate([re.astype(object), im.astype(object)], axis=-1)
        flat = coeffs.reshape(-1)
        for i in range(flat.shape[0]):
            flat[i] = int(flat[i])
        return coeffs

    def decode(self, coeffs: np.ndarray, scale: float, k: int) -> np.ndarray:
        """Centered integer coeffs [..., n] -> real slot values [..., k]."""
        c = np.asarray(coeffs)
        half = self.slots
        cf = c.astype(np.float64)
        z = cf[..., :half] + 1j * cf[..., half:]
        v = self.fft_special(z / scale)
        return v.real[..., :k]
