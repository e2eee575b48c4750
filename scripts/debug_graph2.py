import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from hefl.config import preset
from hefl.fl.client import LocalClient

cfg = preset("config2"); cfg.fl.n_clients = 1
c = LocalClient(cfg, 0, device="cuda:0")
x, y = next(iter(c.loader))
w0 = c.get_weights()
# graphed: replay same batch 50x
for i in range(50):
    loss, logits = c.train_step(x, y)
    if i % 10 == 0:
        torch.cuda.synchronize()
        print(f"step {i} loss={float(loss):.4f}")
torch.cuda.synchronize()
w1 = c.get_weights()
print("wdelta:", (w1 - w0).abs().max().item())
# check grads non-zero after a replay
for j, p in enumerate(c.opt.params):
    g = p.grad
    print(j, tuple(p.shape), "grad_absmax", None if g is None else g.abs().max().item())
