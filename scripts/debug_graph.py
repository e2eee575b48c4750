import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from hefl.ops.adam import FusedAdam
from hefl.ops import functional as Fx

# T1: bare in-place update captured
p = torch.zeros(10, device='cuda')
side = torch.cuda.Stream(); side.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(side):
    p.add_(1.0)
torch.cuda.current_stream().wait_stream(side)
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g, stream=side):
    p.add_(1.0)
g.replay(); g.replay(); torch.cuda.synchronize()
print('T1 (expect 3.0):', p[0].item())

# T2: our linear + backward + step_graphed captured
torch.manual_seed(0)
x = torch.randn(4, 8, device='cuda', dtype=torch.bfloat16)
w = torch.nn.Parameter(torch.randn(6, 8, device='cuda') * 0.1)
opt = FusedAdam([w], lr=0.1)
def eager():
    y = Fx.linear(x, w)
    loss = (y.float() ** 2).sum()
    opt.zero_grad_()
    loss.backward()
    opt.step()
side2 = torch.cuda.Stream(); side2.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(side2):
    for _ in range(3):
        eager()
torch.cuda.current_stream().wait_stream(side2)
opt.prepare_graph_state(x.device)
gid = id(w.grad); gptr = w.grad.data_ptr()
g2 = torch.cuda.CUDAGraph()
with torch.cuda.graph(g2, stream=side2):
    y = Fx.linear(x, w)
    loss = (y.float() ** 2).sum()
    opt.zero_grad_()
    loss.backward()
    opt.step_graphed()
print('T2 grad identity same:', id(w.grad) == gid, 'ptr same:', w.grad.data_ptr() == gptr)
w0 = w.detach().clone()
g2.replay(); torch.cuda.synchronize()
w1 = w.detach().clone()
g2.replay(); torch.cuda.synchronize()
w2 = w.detach().clone()
print('T2 delta after replay1:', (w1 - w0).abs().max().item(),
      'replay2:', (w2 - w1).abs().max().item())
print('T2 grad after replays:', w.grad.abs().max().item(),
      'step_t:', opt._step_t.item(), 'sched:', opt._sched.tolist())
