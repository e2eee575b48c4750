import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from hefl.config import preset
from hefl.fl.client import LocalClient

def run(use_graphs, steps=300):
    cfg = preset("config2")
    cfg.fl.n_clients = 1
    cfg.train.hip_graphs = use_graphs
    c = LocalClient(cfg, 0, device="cuda:0")
    losses = []
    n = 0
    for ep in range(20):
        for x, y in c.loader:
            loss, logits = c.train_step(x, y)
            n += 1
            if n % 23 == 0:
                losses.append(float(loss))
            if n >= steps:
                break
        if n >= steps:
            break
    w = c.get_weights()
    print(f"graphs={use_graphs} losses={[round(l,3) for l in losses]} "
          f"nan_weights={bool(torch.isnan(w).any())} wmax={w.abs().max().item():.3f}")

run(False)
run(True)
