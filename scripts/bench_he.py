"""CKKS primitive microbench: per-op device time for the BASELINE configs."""
import sys, os, time; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from hefl.config import HEConfig
from hefl.he.ckks import CKKSContext, CtxtTensor

def t(fn, iters=20):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        out = fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3, out

cases = [
    ("config2: 222k wts, m=2^13 L2", HEConfig(m=8192, scale_bits=40, q_bits=(60, 40), seed=1), 222_722),
    ("config3: 62k wts,  m=2^14 L3", HEConfig(m=16384, scale_bits=40, q_bits=(60, 40, 40), seed=1), 62_006),
    ("config5: 11.2M wts, m=2^15 L4", HEConfig(m=32768, scale_bits=40, q_bits=(60, 40, 40, 40), seed=1), 11_181_642),
]
for name, cfg, nw in cases:
    ctx = CKKSContext(cfg, device="cuda")
    kp = ctx.keygen()
    vec = torch.randn(nw, device="cuda")
    ms_enc, ct = t(lambda: ctx.encrypt_tensor(vec, kp.pk))
    B = ct.data.shape[0]
    lazy = CtxtTensor(ct.data * 1, ct.scale, ct.count)
    lazy.data.mul_(3)  # fake 3-client lazy sum magnitude
    ms_red, _ = t(lambda: ctx.modreduce_tensor_(lazy))
    ms_mul, sc = t(lambda: ctx.mul_scalar_tensor(ct, 0.125))
    ms_rs, rs = t(lambda: ctx.rescale_tensor(sc))
    ms_dec, _ = t(lambda: ctx.decrypt_tensor(rs, kp.sk))
    mb = ct.data.numel() * 8 / 1e6
    line = (f"{name}: B={B} cts ({mb:.0f} MB) | encrypt {ms_enc:7.2f}ms | "
            f"modreduce {ms_red:6.2f}ms | ct*plain {ms_mul:6.2f}ms | "
            f"rescale {ms_rs:6.2f}ms | decrypt {ms_dec:7.2f}ms")
    if ctx.L >= 3:
        # ct x ct + relinearize (the config #3/#5 aggregation inner op):
        # batched ciphertext tensor times one shared encrypted denominator
        import numpy as np
        from hefl.he.ckks import Ciphertext
        rlk = ctx.relin_keygen(kp.sk)
        denom = ctx.encrypt(ctx.encode(np.full(ctx.slots, 0.125)), kp.pk)
        ctc = Ciphertext(ct.data, ct.scale)
        ms_mc, _ = t(lambda: ctx.mul_ct(ctc, denom, rlk), iters=5)
        line += f" | ct*ct+relin {ms_mc:7.2f}ms"
    print(line)
