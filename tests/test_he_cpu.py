"""CPU tests for the CKKS engine (the numerics oracle for the HIP kernels).

Mirrors the validation the reference has implicitly (notebook accuracy check
+ the commented decrypt probe at FLPyfhelin.py:382) but as real property
tests, per SURVEY.md section 4.
"""
import numpy as np
import pytest
import torch

from hefl.config import HEConfig
from hefl.he.ckks import CKKSContext, CtxtTensor
from hefl.he.ntt_cpu import NttTables, fwd_ntt, inv_ntt, negacyclic_mul_naive
from hefl.he.primes import gen_prime_chain, is_prime


SMALL = HEConfig(m=64, scale_bits=30, q_bits=(50, 30), seed=1)


@pytest.fixture(scope="module")
def ctx():
    return CKKSContext(SMALL)


@pytest.fixture(scope="module")
def keys(ctx):
    return ctx.keygen()


def test_prime_chain_properties():
    for n in (64, 8192):
        primes = gen_prime_chain(n, (60, 40, 40))
        assert len(set(primes)) == 3
        for q in primes:
            assert is_prime(q)
            assert q % (2 * n) == 1
            assert q < 2 ** 60  # lazy all-reduce bound


def test_ntt_roundtrip_and_convolution():
    n = 64
    q = gen_prime_chain(n, (50,))[0]
    tb = NttTables(q, n)
    rng = np.random.default_rng(0)
    a = rng.integers(0, q, size=n).astype(object)
    b = rng.integers(0, q, size=n).astype(object)
    # roundtrip
    assert np.array_equal(inv_ntt(fwd_ntt(a, tb), tb), a)
    # negacyclic convolution == pointwise product in NTT domain
    prod = inv_ntt((fwd_ntt(a, tb) * fwd_ntt(b, tb)) % q, tb)
    ref = negacyclic_mul_naive(a, b, q, n)
    assert np.array_equal(prod, ref)


def test_encrypt_decrypt_roundtrip(ctx, keys):
    v = np.random.default_rng(2).normal(size=ctx.slots)
    ct = ctx.encrypt(ctx.encode(v), keys.pk)
    out = ctx.decode(ctx.decrypt(ct, keys.sk), ctx.slots)
    assert np.abs(out - v).max() < 1e-4


def test_homomorphic_add(ctx, keys):
    rng = np.random.default_rng(3)
    a, b = rng.normal(size=ctx.slots), rng.normal(size=ctx.slots)
    ca = ctx.encrypt(ctx.encode(a), keys.pk)
    cb = ctx.encrypt(ctx.encode(b), keys.pk)
    out = ctx.decode(ctx.decrypt(ctx.add(ca, cb), keys.sk), ctx.slots)
    assert np.abs(out - (a + b)).max() < 1e-4


def test_scalar_mult_and_rescale(ctx, keys):
    v = np.random.default_rng(4).normal(size=ctx.slots)
    ct = ctx.encrypt(ctx.encode(v), keys.pk)
    out_ct = ctx.rescale(ctx.mul_scalar(ct, 0.125))
    assert out_ct.level == ct.level - 1
    out = ctx.decode(ctx.decrypt(out_ct, keys.sk), ctx.slots)
    assert np.abs(out - 0.125 * v).max() < 1e-4


def test_tensor_pack_roundtrip(ctx, keys):
    vec = torch.randn(5 * ctx.slots + 7)  # forces multiple cts + padding
    ct = ctx.encrypt_tensor(vec, keys.pk)
    assert ct.data.shape[0] == 6
    back = ctx.decrypt_tensor(ct, keys.sk)
    assert (back - vec).abs().max().item() < 1e-4


def test_lazy_sum_fedavg_semantics(ctx, keys):
    """The RCCL all-reduce path: int64 SUM of <=8 ciphertexts without
    per-hop reduction, then one modreduce + 1/n scalar mult + rescale."""
    n_clients = 8
    vec = torch.randn(100)
    cts = [ctx.encrypt_tensor(vec * (i + 1), keys.pk) for i in range(n_clients)]
    summed = cts[0].data.clone()
    for c in cts[1:]:
        summed += c.data  # raw int64 add — must not overflow (q < 2**60)
    agg = CtxtTensor(summed, cts[0].scale, cts[0].count)
    ctx.modreduce_tensor_(agg)
    avg = ctx.rescale_tensor(ctx.mul_scalar_tensor(agg, 1.0 / n_clients))
    expect = vec * (sum(range(1, n_clients + 1)) / n_clients)
    out = ctx.decrypt_tensor(avg, keys.sk)
    assert (out - expect).abs().max().item() < 1e-3


def test_mul_ct_relin_rescale():
    cfg = HEConfig(m=64, scale_bits=26, q_bits=(55, 26, 26), seed=0)
    ctx = CKKSContext(cfg)
    kp = ctx.keygen()
    rlk = ctx.relin_keygen(kp.sk)
    rng = np.random.default_rng(5)
    v1, v2 = rng.normal(size=32) * 0.5, rng.normal(size=32) * 0.5
    ct1 = ctx.encrypt(ctx.encode(v1), kp.pk)
    ct2 = ctx.encrypt(ctx.encode(v2), kp.pk)
    out_ct = ctx.rescale(ctx.mul_ct(ct1, ct2, rlk))
    out = ctx.decode(ctx.decrypt(out_ct, kp.sk), 32)
    assert np.abs(out - v1 * v2).max() < 1e-3


def test_rescale_matches_exact_division(ctx, keys):
    """rescale is exact RNS division by q_last with rounding: check
    dec(rescale(ct)) ~= dec(ct)/q_last via the scale bookkeeping."""
    v = np.random.default_rng(6).normal(size=ctx.slots)
    ct = ctx.encrypt(ctx.encode(v), keys.pk)
    r = ctx.rescale(ctx.mul_scalar(ct, 1.0))
    out = ctx.decode(ctx.decrypt(r, keys.sk), ctx.slots)
    assert np.abs(out - v).max() < 1e-4


def test_security_standard_check():
    """BASELINE configs meet the HE-standard 128-bit classical bound
    (chain + special prime)."""
    from hefl.config import preset
    for name in ("config2", "config3", "config4", "config5"):
        cfg = preset(name).he
        ctx = CKKSContext(HEConfig(m=cfg.m, scale_bits=cfg.scale_bits,
                                   q_bits=cfg.q_bits, seed=0))
        assert ctx.secure_128, (name, sum(q.bit_length()
                                          for q in ctx.all_primes))


def test_insecure_params_warn():
    import warnings
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        CKKSContext(HEConfig(m=1024, scale_bits=40, q_bits=(60, 40, 60),
                             seed=0))  # logQP ~ 220 >> 27
        assert any("128-bit" in str(x.message) for x in w)
