"""End-to-end encrypted FL on the GPU kernels: accuracy-parity check
(the reference's only validation mechanism, SURVEY.md section 4 item 1,
run on the real MI355X path: HIP CNN training + CKKS kernels + graphs)."""
import pytest
import torch

from hefl.config import preset
from hefl.fl.sequential import SequentialFL

pytestmark = pytest.mark.gpu


def test_gpu_encrypted_fl_reaches_accuracy():
    torch.manual_seed(0)
    cfg = preset("config2")
    cfg.fl.n_clients = 2
    cfg.fl.samples_per_client = 96
    cfg.fl.test_samples = 96
    cfg.model.n_classes = 2
    cfg.he.seed = 17
    fl = SequentialFL(cfg, device="cuda:0")
    rep = None
    for _ in range(2):
        rep = fl.run_round(epochs=2)
    assert rep.metrics["accuracy"] > 0.8, rep.metrics
    # encrypted aggregation really ran on device
    assert fl.ctx is not None and fl.ctx.device.type == "cuda"


def test_gpu_phase_timer():
    from hefl.utils import PhaseTimer

    t = PhaseTimer()
    x = torch.randn(4096, 4096, device="cuda", dtype=torch.bfloat16)
    with t.phase("matmul-ish"):
        y = (x * 2).sum()
    s = t.summary()
    assert s["matmul-ish"]["device_s"] >= 0
