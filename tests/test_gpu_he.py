"""GPU numerics tests: HIP CKKS kernels vs the exact CPU oracle.

Every test compares hefl._C kernels against the big-integer CPU reference
(hefl/he/ntt_cpu.py) or the CPU CKKSContext on identical inputs.
"""
import numpy as np
import pytest
import torch

import hefl
from hefl.config import HEConfig
from hefl.he.ckks import CKKSContext, CtxtTensor
from hefl.he.ntt_cpu import NttTables, fwd_ntt, inv_ntt
from hefl.he.primes import gen_prime_chain

pytestmark = pytest.mark.gpu


def _tables_to_gpu(q, n):
    tb = NttTables(q, n)
    w = torch.from_numpy(tb.w.astype(np.int64)).cuda()
    winv = torch.from_numpy(tb.winv.astype(np.int64)).cuda()

    def shoup(v):
        return torch.from_numpy(
            ((v.cpu().numpy().astype(object) << 64) // q)
            .astype(np.uint64).astype(np.int64)).cuda()

    ninv = tb.n_inv
    ninvsh = (ninv << 64) // q
    return tb, w, shoup(w), winv, shoup(winv), ninv, int(np.int64(np.uint64(ninvsh)))


@pytest.mark.parametrize("n", [64, 1024, 8192, 16384, 32768])
def test_ntt_matches_cpu_oracle(n):
    C = hefl.load_extension()
    q = gen_prime_chain(n, (60,))[0]
    tb, w, wsh, winv, winvsh, ninv, ninvsh = _tables_to_gpu(q, n)
    rng = np.random.default_rng(0)
    rows = 3
    a = rng.integers(0, q, size=(rows, n), dtype=np.int64)
    ref = fwd_ntt(a.astype(object), tb).astype(np.int64)

    x = torch.from_numpy(a).cuda()
    C.ntt_batch(x, w, wsh, q)
    assert np.array_equal(x.cpu().numpy(), ref), "forward NTT mismatch"

    C.intt_batch(x, winv, winvsh, q, ninv, ninvsh)
    assert np.array_equal(x.cpu().numpy(), a), "inverse NTT roundtrip mismatch"


def test_pointwise_modops():
    C = hefl.load_extension()
    q = gen_prime_chain(1024, (60,))[0]
    rng = np.random.default_rng(1)
    a = rng.integers(0, q, size=(4, 1024), dtype=np.int64)
    b = rng.integers(0, q, size=(4, 1024), dtype=np.int64)
    ag, bg = torch.from_numpy(a).cuda(), torch.from_numpy(b).cuda()

    mm = C.modmul(ag, bg, q).cpu().numpy()
    ref = (a.astype(object) * b.astype(object)) % q
    assert np.array_equal(mm, ref.astype(np.int64))

    s = int(rng.integers(1, q))
    ms = C.modmul_scalar(ag, s, q).cpu().numpy()
    assert np.array_equal(ms, ((a.astype(object) * s) % q).astype(np.int64))

    ad = C.modadd(ag, bg, q).cpu().numpy()
    assert np.array_equal(ad, ((a.astype(object) + b) % q).astype(np.int64))

    su = C.modsub(ag, bg, q).cpu().numpy()
    assert np.array_equal(su, ((a.astype(object) - b) % q).astype(np.int64))


def test_modreduce_lazy_sum():
    C = hefl.load_extension()
    qs_list = gen_prime_chain(256, (60, 40))
    rng = np.random.default_rng(2)
    # 8-client lazy sums: values up to 8 * q < 2^63
    x = np.stack([rng.integers(0, 8 * q, size=(5, 256), dtype=np.int64)
                  for q in qs_list], axis=1)  # [5, L, 256]
    ref = np.stack([x[:, i].astype(object) % q
                    for i, q in enumerate(qs_list)], axis=1).astype(np.int64)
    xg = torch.from_numpy(x.copy()).cuda()
    C.modreduce_(xg, torch.tensor(qs_list, dtype=torch.int64).cuda())
    assert np.array_equal(xg.cpu().numpy(), ref)


def test_gpu_context_encrypt_decrypt_matches_cpu():
    """GPU CKKSContext must round-trip and agree with CPU on decode."""
    cfg = HEConfig(m=8192, scale_bits=40, q_bits=(60, 40), seed=7)
    gpu = CKKSContext(cfg, device="cuda")
    kp = gpu.keygen()
    vec = torch.randn(10000)
    ct = gpu.encrypt_tensor(vec, kp.pk)
    assert ct.data.is_cuda
    back = gpu.decrypt_tensor(ct, kp.sk).cpu()
    assert (back - vec).abs().max().item() < 1e-3

    # homomorphic FedAvg semantics on GPU
    ct2 = gpu.encrypt_tensor(vec * 3, kp.pk)
    lazy = CtxtTensor(ct.data + ct2.data, ct.scale, ct.count)
    gpu.modreduce_tensor_(lazy)
    avg = gpu.rescale_tensor(gpu.mul_scalar_tensor(lazy, 0.5))
    out = gpu.decrypt_tensor(avg, kp.sk).cpu()
    assert (out - 2 * vec).abs().max().item() < 1e-2


def test_gpu_vs_cpu_identical_ntt_path():
    """Same seed => GPU and CPU contexts produce identical ciphertext ints."""
    cfg = HEConfig(m=1024, scale_bits=30, q_bits=(50, 30), seed=3)
    cpu = CKKSContext(cfg, device="cpu")
    gpu = CKKSContext(cfg, device="cuda")
    kp_c = cpu.keygen()
    kp_g = gpu.keygen()
    assert torch.equal(kp_c.sk, kp_g.sk.cpu())
    assert torch.equal(kp_c.pk, kp_g.pk.cpu())
    v = np.linspace(-1, 1, cfg.m // 2)
    pt_c = cpu.encode(v)
    pt_g = gpu.encode(v)
    # GPU encode runs the special FFT in torch complex128 on device; FMA
    # contraction can flip a rounding at the int boundary, so compare the
    # DECODED values (and allow <=1 ulp-of-int coefficient differences)
    diff = (pt_c.data - pt_g.data.cpu()).abs()
    q0 = cpu.primes[0]
    assert ((diff == 0) | (diff == 1) | (diff == q0 - 1)).float().mean() > 0.99
    out_c = cpu.decode(pt_c, 16)
    out_g = gpu.decode(pt_g, 16)
    og = out_g.cpu().numpy() if torch.is_tensor(out_g) else out_g
    assert np.abs(out_c - og).max() < 1e-6


@pytest.mark.parametrize("n", [1024, 32768])
def test_fused_multilimb_ops_match_per_limb(n):
    """ntt_limbs / modmul_limbs / modmul_scalar_limbs must equal the
    per-limb kernels on identical inputs (n=2^15 exercises the fused
    radix-4 cross-block stage)."""
    cfg = HEConfig(m=n, scale_bits=30, q_bits=(50, 40, 30), seed=9)
    ctx = CKKSContext(cfg, device="cuda")
    be = ctx.backend
    L = 3
    rng = np.random.default_rng(4)
    x = torch.from_numpy(np.stack(
        [rng.integers(0, ctx.primes[i], size=(5, n), dtype=np.int64)
         for i in range(L)], axis=1)).cuda()  # [5, L, n]

    fused = be.ntt_all(x)
    per = torch.stack([be.ntt(x[:, i, :], i) for i in range(L)], dim=1)
    assert torch.equal(fused, per)

    back = be.ntt_all(fused, inverse=True)
    assert torch.equal(back, x)

    b = torch.from_numpy(np.stack(
        [rng.integers(0, ctx.primes[i], size=(n,), dtype=np.int64)
         for i in range(L)], axis=0)).cuda()  # [L, n]
    mm = be.modmul_limbs(x, b)
    per = torch.stack(
        [be.modmul(x[:, i, :].contiguous(),
                   b[i].expand(5, n).contiguous(), i) for i in range(L)], dim=1)
    assert torch.equal(mm, per)

    scalars = [123456789, 987654321, 55555]
    ms = be.modmul_scalar_limbs(x, scalars)
    per = torch.stack([be.modmul_scalar(x[:, i, :].contiguous(),
                                        scalars[i], i) for i in range(L)], dim=1)
    assert torch.equal(ms, per)


def test_device_noise_sampler_distribution():
    """cbd21 must be a centered binomial with sigma = sqrt(21/2) ~ 3.24
    (regression: an unsigned popcount subtraction once wrapped negatives to
    ~2^32, silently destroying every ciphertext)."""
    import hefl
    C = hefl.load_extension()
    g = torch.Generator(device="cuda")
    g.manual_seed(3)
    bits = torch.randint(-(2 ** 63), 2 ** 63 - 1, (1_000_000,), generator=g,
                         device="cuda", dtype=torch.int64)
    e = C.cbd21(bits).float()
    assert abs(e.mean().item()) < 0.02
    assert abs(e.std().item() - (21 / 2) ** 0.5) < 0.05
    assert e.abs().max().item() <= 21


@pytest.mark.gpu
def test_synth_batch_distribution():
    """Fused data-gen kernel: logit(x) must equal 0.6*template + 0.4*N(0,1)
    in distribution (mean residual ~0, std ~0.4)."""
    from hefl.data.synthetic import SyntheticMedicalImages

    ds = SyntheticMedicalImages(512, (28, 28, 1), 4, seed=3, device="cuda",
                                dtype=torch.bfloat16)
    idx = torch.arange(512, device="cuda")
    x, y = ds.batch(idx)
    assert x.dtype == torch.bfloat16 and x.shape == (512, 28, 28, 1)
    xf = x.float().clamp(1e-4, 1 - 1e-4)
    logit = torch.log(xf / (1 - xf))
    resid = logit - 0.6 * ds.templates[y]
    assert abs(resid.mean().item()) < 0.02, resid.mean().item()
    assert abs(resid.std().item() - 0.4) < 0.05, resid.std().item()
    # different calls draw fresh noise
    x2, _ = ds.batch(idx)
    assert not torch.equal(x, x2)


@pytest.mark.parametrize("m", [256, 8192, 16384, 32768])
def test_hip_fft_encode_decode_matches_oracle(m):
    """Hand-written special-FFT kernels (hefl/csrc/fft.hip) vs the CPU
    oracle encoder — the VERDICT r1 item 5 replacement for the torch
    complex128 at::native path. m=32768 (slots 2^14) exercises the
    global-stage split (slots > the 8192-slot LDS block)."""
    import hefl
    from hefl.he.encoder import Encoder
    C = hefl.load_extension()
    enc = Encoder(m)
    slots = m // 2
    rng = np.random.default_rng(5)
    vals = rng.standard_normal((3, slots))
    scale = 2.0 ** 40
    ref = enc.encode(vals, scale)              # CPU oracle
    tw_enc, tw_dec = enc._hip_tables("cuda")
    got = C.fft_encode(torch.from_numpy(vals).cuda(), tw_enc, scale).cpu()
    diff = (got - torch.from_numpy(np.asarray(ref, dtype=np.int64))).abs()
    # identical butterfly order; FMA contraction may flip an int rounding
    assert (diff <= 1).all(), diff.max()
    assert (diff == 0).float().mean() > 0.99
    back = C.fft_decode(got.cuda(), tw_dec, scale, slots).cpu().numpy()
    ref_back = enc.decode(np.asarray(ref), scale, slots)
    assert np.abs(back - ref_back).max() < 1e-5
    assert np.abs(back - vals).max() < 1e-4


def test_gpu_synth_affine_augment():
    """In-kernel zoom/shear/flip augmentation: deterministic per seed,
    range-preserving, and actually transforming (vs the plain path)."""
    import hefl
    from hefl.data.synthetic import SyntheticMedicalImages
    C = hefl.load_extension()
    ds = SyntheticMedicalImages(64, (32, 32, 3), 2, seed=9, device="cuda",
                                dtype=torch.bfloat16)
    idx = torch.arange(16)
    base, _ = ds.batch(idx)
    ds2 = SyntheticMedicalImages(64, (32, 32, 3), 2, seed=9, device="cuda",
                                 dtype=torch.bfloat16)
    aug, y = ds2.batch(idx, affine=(0.2, 0.2, True))
    ds3 = SyntheticMedicalImages(64, (32, 32, 3), 2, seed=9, device="cuda",
                                 dtype=torch.bfloat16)
    aug2, _ = ds3.batch(idx, affine=(0.2, 0.2, True))
    assert torch.equal(aug, aug2)
    assert not torch.equal(aug, base)
    af = aug.float()
    assert af.min() >= 0 and af.max() <= 1
    # same noise model: moments match the plain batch closely
    assert abs(af.mean().item() - base.float().mean().item()) < 0.03


def test_gpu_mul_ct_relin_matches_cpu_context():
    """Fused ct_mul + ks_inner + mod-down (one-launch key-switch) against
    the exact CPU context with identical keys/noise: decrypted products
    must agree to CKKS noise, and the GPU path's ciphertext ints must stay
    valid residues."""
    cfg = HEConfig(m=512, scale_bits=26, q_bits=(55, 26, 26), seed=21)
    cpu = CKKSContext(cfg, device="cpu")
    gpu = CKKSContext(cfg, device="cuda")
    kp_c = cpu.keygen()
    kp_g = gpu.keygen()
    assert torch.equal(kp_c.sk, kp_g.sk.cpu())
    rlk_c = cpu.relin_keygen(kp_c.sk)
    rlk_g = gpu.relin_keygen(kp_g.sk)
    assert torch.equal(rlk_c, rlk_g.cpu())  # same host-sampled key material
    va = np.linspace(-1, 1, cfg.m // 2)
    vb = np.linspace(0.5, -0.5, cfg.m // 2)
    ct_ac = cpu.encrypt(cpu.encode(va), kp_c.pk)
    ct_bc = cpu.encrypt(cpu.encode(vb), kp_c.pk)
    ct_ag = gpu.encrypt(gpu.encode(va), kp_g.pk)
    ct_bg = gpu.encrypt(gpu.encode(vb), kp_g.pk)
    prod_c = cpu.rescale(cpu.mul_ct(ct_ac, ct_bc, rlk_c))
    prod_g = gpu.rescale(gpu.mul_ct(ct_ag, ct_bg, rlk_g))
    # residues valid
    for i in range(prod_g.level):
        assert prod_g.data[..., i, :].max().item() < gpu.primes[i]
        assert prod_g.data[..., i, :].min().item() >= 0
    out_c = cpu.decode(cpu.decrypt(prod_c, kp_c.sk), 64)
    out_g = gpu.decode(gpu.decrypt(prod_g, kp_g.sk), 64)
    og = out_g.cpu().numpy() if torch.is_tensor(out_g) else out_g
    ref = (va * vb)[:64]
    assert np.abs(out_c - ref).max() < 1e-2
    assert np.abs(og - ref).max() < 1e-2
    # same-seed contexts produce numerically equal results end to end
    assert np.abs(og - out_c).max() < 1e-3
