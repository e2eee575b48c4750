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


# ---------------------------------------------------------------------------
# Torch port of the special FFT pair: runs on the context device (MI355X) in
# complex128. The numpy path above stays as the CPU oracle; this path exists
# because host-side encode/decode of a ResNet-sized weight vector (hundreds
# of ciphertexts x 2^14 slots) costs seconds in numpy and ~1 ms on device.
# ---------------------------------------------------------------------------

import torch as _torch


class TorchEncoderMixin:
    def _torch_tables(self, device):
        cache = getattr(self, "_tcache", None)
        if cache is None:
            cache = self._tcache = {}
        key = str(device)
        if key in cache:
            return cache[key]
        perm = _torch.tensor(
            [int(bit_reverse(i, self.slots.bit_length() - 1))
             for i in range(self.slots)], dtype=_torch.long, device=device)
        fwd, inv = [], []
        size = self.slots
        length = 2
        while length <= size:
            lenh, lenq = length // 2, length * 4
            idx = [(self.rot[j] % lenq) * (self.M // lenq) for j in range(lenh)]
            fwd.append(_torch.from_numpy(self.ksi[idx]).to(device))
            length *= 2
        length = size
        while length >= 2:
            lenh, lenq = length // 2, length * 4
            idx = [(lenq - (self.rot[j] % lenq)) * (self.M // lenq)
                   for j in range(lenh)]
            inv.append(_torch.from_numpy(self.ksi[idx]).to(device))
            length //= 2
        cache[key] = (perm, fwd, inv)
        return cache[key]

    def fft_special_torch(self, v: "_torch.Tensor") -> "_torch.Tensor":
        """Decode direction. v: complex128 [..., slots] on device."""
        perm, fwd, _ = self._torch_tables(v.device)
        v = v.index_select(-1, perm)
        size = self.slots
        length = 2
        si = 0
        while length <= size:
            lenh = length // 2
            w = fwd[si]
            lead = v.shape[:-1]
            v = v.reshape(*lead, size // length, 2, lenh)
            u = v[..., 0, :]
            t = v[..., 1, :] * w
            v = _torch.stack([u + t, u - t], dim=-2).reshape(*lead, size)
            length *= 2
            si += 1
        return v

    def fft_special_inv_torch(self, v: "_torch.Tensor") -> "_torch.Tensor":
        """Encode direction. v: complex128 [..., slots] on device."""
        perm, _, inv = self._torch_tables(v.device)
        size = self.slots
        length = size
        si = 0
        while length >= 2:
            lenh = length // 2
            w = inv[si]
            lead = v.shape[:-1]
            v = v.reshape(*lead, size // length, 2, lenh)
            u = v[..., 0, :] + v[..., 1, :]
            t = (v[..., 0, :] - v[..., 1, :]) * w
            v = _torch.stack([u, t], dim=-2).reshape(*lead, size)
            length //= 2
            si += 1
        v = v.index_select(-1, perm)
        return v / size

    def _hip_tables(self, device):
        """Stage-major twiddle tensors for the hand-written HIP special-FFT
        kernels (hefl/csrc/fft.hip): DIF order (length = slots down to 2)
        for encode, DIT order (2 up to slots) for decode. f64 interleaved
        (re, im), cached per device."""
        cache = getattr(self, "_hcache", None)
        if cache is None:
            cache = self._hcache = {}
        key = str(device)
        if key in cache:
            return cache[key]
        import numpy as _np
        enc, dec = [], []
        size = self.slots
        length = size
        while length >= 2:
            lenh, lenq = length // 2, length * 4
            idx = [(lenq - (self.rot[j] % lenq)) * (self.M // lenq)
                   for j in range(lenh)]
            enc.append(self.ksi[idx])
            length //= 2
        length = 2
        while length <= size:
            lenh, lenq = length // 2, length * 4
            idx = [(self.rot[j] % lenq) * (self.M // lenq)
                   for j in range(lenh)]
            dec.append(self.ksi[idx])
            length *= 2
        def pack(stages):
            z = _np.ascontiguousarray(_np.concatenate(stages))
            return _torch.from_numpy(z.view(_np.float64).reshape(-1, 2)).to(device)
        cache[key] = (pack(enc), pack(dec))
        return cache[key]

    def encode_torch(self, vals: "_torch.Tensor", scale: float) -> "_torch.Tensor":
        """vals: real [..., k<=slots] on device -> int64 coeffs [..., n]
        (centered; exact for |coeff| < 2^52 — weights are O(1) and scales
        <= 2^45, so coefficients sit far below the f64 integer range; the
        CPU oracle path keeps the explicit overflow guard)."""
        vals = vals.to(_torch.float64)
        pad = self.slots - vals.shape[-1]
        if pad < 0:
            raise ValueError("too many values for slot count")
        if pad:
            vals = _torch.nn.functional.pad(vals, (0, pad))
        if vals.is_cuda:
            import hefl
            tw_enc, _ = self._hip_tables(vals.device)
            return hefl.load_extension().fft_encode(vals.contiguous(), tw_enc,
                                                    float(scale))
        z = self.fft_special_inv_torch(vals.to(_torch.complex128))
        re = _torch.round(z.real * scale)
        im = _torch.round(z.imag * scale)
        peak = max(re.abs().max().item(), im.abs().max().item())
        if peak >= 2.0 ** 52:
            raise OverflowError("encode overflow: scale too large for f64 path")
        return _torch.cat([re, im], dim=-1).to(_torch.int64)

    def decode_torch(self, coeffs: "_torch.Tensor", scale: float,
                     k: int) -> "_torch.Tensor":
        """Centered int64 coeffs [..., n] on device -> float32 [..., k]."""
        if coeffs.is_cuda:
            import hefl
            _, tw_dec = self._hip_tables(coeffs.device)
            return hefl.load_extension().fft_decode(coeffs.contiguous(),
                                                    tw_dec, float(scale), k)
        half = self.slots
        cf = coeffs.to(_torch.float64)
        z = _torch.complex(cf[..., :half], cf[..., half:]) / scale
        v = self.fft_special_torch(z)
        return v.real[..., :k].to(_torch.float32)


# graft the torch methods onto Encoder (mixin-by-assignment; __bases__
# reassignment is not allowed on classes deriving from object directly)
for _name in ("_torch_tables", "_hip_tables", "fft_special_torch",
              "fft_special_inv_torch", "encode_torch", "decode_torch"):
    setattr(Encoder, _name, getattr(TorchEncoderMixin, _name))
