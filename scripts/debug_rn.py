import sys, os, time; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from hefl.config import preset
from hefl.fl.client import LocalClient

cfg = preset("config5"); cfg.fl.n_clients = 1
c = LocalClient(cfg, 0, device="cuda:0")
batches = list(c.loader)[:6]
x, y = batches[0]
# first call triggers capture
t0 = time.perf_counter(); c.train_step(x, y); torch.cuda.synchronize()
print(f"capture+1st: {time.perf_counter()-t0:.3f}s graphs={len(c._graphs)}")
for i in range(3):
    t0 = time.perf_counter()
    for x, y in batches:
        c.train_step(x, y)
    torch.cuda.synchronize()
    print(f"6 steps: {(time.perf_counter()-t0)*1000:.1f}ms -> {(time.perf_counter()-t0)/6*1000:.2f} ms/step graphs={len(c._graphs)}")
# data loading cost alone
t0 = time.perf_counter()
for _ in range(3):
    for x, y in c.loader: pass
torch.cuda.synchronize()
print(f"loader alone: {(time.perf_counter()-t0)/3*1000:.1f} ms/epoch")
