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


def test_rescale_level_exhaustion_raises(ctx, keys):
    """A ciphertext at the last level cannot rescale again — the engine must
    say so instead of producing garbage (the reference's Pyfhel would abort
    deep in SEAL)."""
    v = np.random.default_rng(7).normal(size=ctx.slots)
    ct = ctx.encrypt(ctx.encode(v), keys.pk)
    r = ctx.rescale(ctx.mul_scalar(ct, 0.5))  # L=2 -> L=1
    assert r.level == 1
    with pytest.raises(Exception):
        ctx.rescale(ctx.mul_scalar(r, 0.5))   # no level left


def test_additive_noise_growth_bounded(ctx, keys):
    """Summing 8 independently-encrypted copies (the lazy all-reduce bound)
    keeps decryption error << one slot-value ulp."""
    rng = np.random.default_rng(8)
    v = rng.normal(size=ctx.slots)
    acc = None
    for _ in range(8):
        ct = ctx.encrypt(ctx.encode(v), keys.pk)
        acc = ct if acc is None else ctx.add(acc, ct)
    out = ctx.decode(ctx.decrypt(acc, keys.sk), ctx.slots)
    assert np.abs(out - 8 * v).max() < 1e-3


def test_ciphertexts_differ_across_ranks_same_plaintext():
    """After reseed(rank), two ranks encrypting the same vector under the
    same shared-seed keys must produce DIFFERENT ciphertexts (client
    privacy), while both still decrypt correctly."""
    cfgs = [HEConfig(m=64, scale_bits=30, q_bits=(50, 30), seed=11)
            for _ in range(2)]
    ctxs = [CKKSContext(c) for c in cfgs]
    kps = [c.keygen() for c in ctxs]
    assert torch.equal(kps[0].pk, kps[1].pk)  # shared-seed keygen
    for rank, c in enumerate(ctxs):
        c.reseed(rank)
    v = np.random.default_rng(9).normal(size=ctxs[0].slots)
    c0 = ctxs[0].encrypt(ctxs[0].encode(v), kps[0].pk)
    c1 = ctxs[1].encrypt(ctxs[1].encode(v), kps[1].pk)
    assert not torch.equal(c0.data, c1.data)
    for c, ct, kp in ((ctxs[0], c0, kps[0]), (ctxs[1], c1, kps[1])):
        out = c.decode(c.decrypt(ct, kp.sk), c.slots)
        assert np.abs(out - v).max() < 1e-4


def test_encrypt_tensor_empty_and_exact_slot_fit(ctx, keys):
    """Boundary shapes: exactly slots-many values (no padding ct) and a
    1-element vector."""
    vec = torch.randn(ctx.slots)
    ct = ctx.encrypt_tensor(vec, keys.pk)
    assert ct.data.shape[0] == 1
    assert (ctx.decrypt_tensor(ct, keys.sk) - vec).abs().max() < 1e-4
    one = torch.tensor([2.5])
    ct1 = ctx.encrypt_tensor(one, keys.pk)
    assert abs(float(ctx.decrypt_tensor(ct1, keys.sk)[0]) - 2.5) < 1e-4


def test_error_paths_raise_clearly():
    """Misuse raises descriptive errors instead of crashing deep in the
    engine (the reference's Pyfhel aborts inside SEAL on most of these)."""
    from hefl.he.pyfhel_compat import Pyfhel

    he = Pyfhel()
    with pytest.raises(ValueError, match="public key"):
        he.encryptFrac(1.0)
    he.contextGen(m=64, q_bits=(50, 30), scale_bits=30, seed=0)
    with pytest.raises(ValueError, match="public key"):
        he.encryptFrac(1.0)
    with pytest.raises(ValueError, match="keyGen"):
        he.relinKeyGen()

    cfg = HEConfig(m=64, scale_bits=26, q_bits=(55, 26, 26), seed=0)
    c = CKKSContext(cfg)
    kp = c.keygen()
    ct = c.encrypt(c.encode(np.ones(32)), kp.pk)
    with pytest.raises(ValueError, match="relinearization"):
        c.mul_ct(ct, ct, None)

    from hefl.config import preset
    with pytest.raises(KeyError, match="unknown preset"):
        preset("nope")
    with pytest.raises(ValueError, match="power of two"):
        CKKSContext(HEConfig(m=100, scale_bits=30, q_bits=(50, 30), seed=0))
    with pytest.raises(ValueError, match="at least one"):
        CKKSContext(HEConfig(m=64, scale_bits=30, q_bits=(), seed=0))
