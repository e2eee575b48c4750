"""Per-shape conv kernel microbench (hipEvent timing, 200 iters)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import hefl
C = hefl.load_extension()

def bench(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    e0.record()
    for _ in range(iters):
        fn()
    e1.record()
    torch.cuda.synchronize()
    return e0.elapsed_time(e1) * 1000 / iters  # us

shapes = [
    ("cnn2-conv1-28x28-c1", 32, 28, 28, 1, 16, 3, 1, 0),
    ("resnet3x3-64c-32x32", 32, 32, 32, 64, 64, 3, 1, 1),
    ("resnet3x3-128c-16x16", 32, 16, 16, 128, 128, 3, 1, 1),
    ("resnet3x3-512c-4x4", 32, 4, 4, 512, 512, 3, 1, 1),
    ("stem7x7-s2-128x128", 32, 128, 128, 3, 64, 7, 2, 3),
    ("cnn4-c2-111x111", 32, 111, 111, 32, 64, 3, 1, 0),
    ("refcnn6-c2-125x125-k32", 32, 125, 125, 32, 32, 3, 1, 0),
    ("refcnn6-c4-28x28-k64", 32, 28, 28, 32, 64, 3, 1, 0),
    ("refcnn6-c1pad-256x256", 32, 256, 256, 8, 32, 3, 1, 0),
]
for name, N, H, W, Cin, Cout, k, st, pad in shapes:
    x = torch.randn(N, H, W, Cin, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(Cout, k, k, Cin, device="cuda", dtype=torch.bfloat16) * 0.1
    b = torch.randn(Cout, device="cuda", dtype=torch.float32)
    OH = (H + 2*pad - k)//st + 1
    OW = (W + 2*pad - k)//st + 1
    dy = torch.randn(N, OH, OW, Cout, device="cuda", dtype=torch.bfloat16)
    flop = 2.0 * N*OH*OW * Cout * k*k*Cin
    t_f = bench(lambda: C.conv2d_fwd(x, w, b, st, True, pad))
    t_d = bench(lambda: C.conv2d_dgrad(dy, w, st, H, W, pad))
    t_w = bench(lambda: C.conv2d_wgrad(dy, x, st, k, k, pad))
    print(f"{name:24s} fwd {t_f:7.1f}us {flop/t_f/1e6:6.1f}TF | "
          f"dgrad {t_d:7.1f}us {flop/t_d/1e6:6.1f}TF | "
          f"wgrad {t_w:7.1f}us {flop/t_w/1e6:6.1f}TF")
