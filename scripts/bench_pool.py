"""Trunk pool kernel microbench: maxpool2x2 fwd + fused pool/relu/bias bwd
per layer shape, with achieved GB/s vs the traffic model."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import hefl
C = hefl.load_extension()

def bench(fn, iters=100):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    e0.record()
    for _ in range(iters):
        fn()
    e1.record()
    torch.cuda.synchronize()
    return e0.elapsed_time(e1) * 1000 / iters  # us

shapes = [  # (name, N, H, W, K) = conv OUTPUT entering the pool
    ("cnn4-l1-222", 32, 222, 222, 32),
    ("cnn4-l2-109", 32, 109, 109, 64),
    ("refcnn6-l1-254", 32, 254, 254, 32),
    ("refcnn6-l2-123", 32, 123, 123, 32),
    ("cnn2-l1-26", 32, 26, 26, 16),
]
for name, N, H, W, K in shapes:
    y = torch.randn(N, H, W, K, device="cuda", dtype=torch.bfloat16).relu()
    p, idx = C.maxpool2x2_fwd(y)
    dy = torch.randn_like(p)
    OH, OW = p.shape[1], p.shape[2]
    t_f = bench(lambda: C.maxpool2x2_fwd(y))
    t_b = bench(lambda: C.pool_relu_bias_bwd(dy, idx, p, H, W))
    fwd_bytes = (N*H*W*K*2 + N*OH*OW*K*3)              # read y, write p+idx
    bwd_bytes = (N*OH*OW*K*5 + N*H*W*K*2)              # dy+p (2B) + idx (1B), write dym
    print(f"{name:16s} fwd {t_f:7.1f}us {fwd_bytes/t_f/1e3:6.0f}GB/s | "
          f"bwd {t_b:7.1f}us {bwd_bytes/t_b/1e3:6.0f}GB/s")
