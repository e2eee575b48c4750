import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import hefl
C = hefl.load_extension()
g = torch.Generator(device="cuda"); g.manual_seed(1)
bits = torch.randint(-(2**63), 2**63 - 1, (1000000,), generator=g, device="cuda", dtype=torch.int64)
print("bits sample:", bits[:4].tolist())
print("bits uniform check: frac negative =", (bits < 0).float().mean().item())
e = C.cbd21(bits)
print("e mean", e.float().mean().item(), "std", e.float().std().item(),
      "min", e.min().item(), "max", e.max().item())
# expected: mean ~0, std ~ sqrt(21/2)=3.24, |e|<=21
from hefl.config import HEConfig
from hefl.he.ckks import CKKSContext
cfg = HEConfig(m=8192, scale_bits=40, q_bits=(60, 40), seed=7)
ctx = CKKSContext(cfg, device="cuda")
kp = ctx.keygen()
vec = torch.randn(10000)
ct = ctx.encrypt_tensor(vec, kp.pk)
back = ctx.decrypt_tensor(ct, kp.sk).cpu()
print("roundtrip err", (back - vec).abs().max().item())
u = ctx._sample_ternary((4, 8192))
print("ternary uniq", u.unique().tolist())
