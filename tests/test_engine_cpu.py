"""CPU-side tests for the execution-engine bookkeeping that the GPU epoch
graphs rely on (table shapes/offsets, loader epoch ordering, flat-vector
round trips). The GPU behavior itself is covered in test_gpu_*."""
import numpy as np
import pytest
import torch

from hefl.data.synthetic import ClientLoader, SyntheticMedicalImages
from hefl.fl.weights import flat_params, load_flat_params
from hefl.models import CNN2
from hefl.ops.adam import FusedAdam


def test_epoch_order_deterministic_and_complete():
    ds = SyntheticMedicalImages(64, (8, 8, 1), 2, seed=0)
    idx = torch.arange(10, 42)
    l1 = ClientLoader(ds, idx, 8, seed=5)
    l2 = ClientLoader(ds, idx, 8, seed=5)
    o1a, o1b = l1.epoch_order(), l1.epoch_order()
    o2a = l2.epoch_order()
    # same (seed, epoch) -> same order; next epoch differs; always a
    # permutation of the shard
    assert torch.equal(o1a, o2a)
    assert not torch.equal(o1a, o1b)
    assert torch.equal(o1a.sort().values, idx)
    # iterating consumes epochs from the same sequence
    l3 = ClientLoader(ds, idx, 8, seed=5)
    batches = [y for _, y in l3]
    assert sum(b.numel() for b in batches) == 32


def test_adam_mt_table_bookkeeping():
    m = CNN2((28, 28, 1), 2, seed=0)
    opt = FusedAdam(m.parameters(), lr=1e-3)
    meta, sizes = opt._mt_shared()
    n_params = len(opt.params)
    assert sizes.shape[0] == n_params
    assert int(sizes.sum()) == sum(p.numel() for p in opt.params)
    # chunk table covers every tensor in MT_CHUNK steps
    expect_chunks = sum((p.numel() + opt._MT_CHUNK - 1) // opt._MT_CHUNK
                        for p in opt.params)
    assert meta.shape == (expect_chunks, 2)
    # per-step shell: empty pointer table of the right shape, shared meta
    shell = opt.alloc_mt_shell()
    assert shell["ptrs"].shape == (n_params, 5)
    assert shell["meta"] is meta and shell["n"] == expect_chunks
    # rows snapshot matches live pointers
    for p in opt.params:
        p.grad = torch.zeros_like(p)
    rows = opt.current_ptr_rows()
    assert len(rows) == n_params
    for r, p, mm, vv in zip(rows, opt.params, opt.m, opt.v):
        assert r[0] == p.data.data_ptr() and r[1] == p.grad.data_ptr()
        assert r[2] == mm.data_ptr() and r[3] == vv.data_ptr()
    shell2 = opt.alloc_mt_shell()
    opt.fill_mt_shell(shell2, rows)
    assert int(shell2["ptrs"][0, 0]) == rows[0][0]


def test_flat_params_includes_buffers_roundtrip():
    from hefl.models.resnet import ResNet18
    m = ResNet18((32, 32, 3), 2, seed=1)
    # perturb running stats so the buffer part of the vector is non-trivial
    with torch.no_grad():
        for b in m.buffers():
            if b.is_floating_point():
                b.add_(torch.randn_like(b) * 0.1)
    vec = flat_params(m)
    n_param = sum(p.numel() for p in m.parameters())
    n_buf = sum(b.numel() for b in m.buffers() if b.is_floating_point())
    assert vec.numel() == n_param + n_buf
    m2 = ResNet18((32, 32, 3), 2, seed=99)
    load_flat_params(m2, vec)
    for a, b in zip(m.parameters(), m2.parameters()):
        assert torch.equal(a, b)
    for a, b in zip(m.buffers(), m2.buffers()):
        if a.is_floating_point():
            assert torch.equal(a, b)


def test_presets_match_baseline_json():
    import json, os
    from hefl.config import preset
    path = os.path.join(os.path.dirname(__file__), "..", "BASELINE.json")
    if not os.path.exists(path):
        pytest.skip("BASELINE.json not present")
    with open(path) as f:
        base = json.load(f)
    # headline config named by BASELINE must exist and be encrypted CKKS
    cfg = preset("config2")
    assert cfg.fl.encrypted and cfg.he.m == 8192
    assert cfg.train.local_epochs == 10 and cfg.train.batch_size == 32
    assert cfg.fl.samples_per_client == 720
    assert "metric" in base or "benchmarks" in base or len(base) > 0


def test_optimizer_resume_syncs_device_schedule_buffers():
    """Checkpoint resume on a graphed client must reach the DEVICE-side
    step/lr buffers the captured Adam kernels read (regression: raw
    attribute writes left them stale)."""
    m = CNN2((28, 28, 1), 2, seed=0)
    opt = FusedAdam(m.parameters(), lr=1e-3, decay=1e-4)
    opt.prepare_graph_state(torch.device("cpu"))  # creates _step_t/_hyper
    opt.step_count = 7
    sd = {"step": 41, "lr": 5e-4, "decay": 2e-4,
          "m": [t.clone() for t in opt.m], "v": [t.clone() for t in opt.v]}
    opt.load_state_dict(sd)
    assert opt.step_count == 41 and int(opt._step_t.item()) == 41
    assert abs(float(opt._hyper[0]) - 5e-4) < 1e-9
    assert abs(float(opt._hyper[1]) - 2e-4) < 1e-9
    # checkpoint round-trip path hits the same buffers
    from hefl.fl.checkpoint import load_round_state, save_round_state
    import tempfile, os
    opt.set_step(99)
    opt.set_lr(3e-4)
    with tempfile.TemporaryDirectory() as d:
        p = os.path.join(d, "r.pt")
        save_round_state(p, m, opt, round_idx=1)
        opt2 = FusedAdam(CNN2((28, 28, 1), 2, seed=1).parameters(), lr=9e-9)
        opt2.prepare_graph_state(torch.device("cpu"))
        load_round_state(p, CNN2((28, 28, 1), 2, seed=1), opt2)
        assert int(opt2._step_t.item()) == 99
        assert abs(float(opt2._hyper[0]) - 3e-4) < 1e-9
