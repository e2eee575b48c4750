"""Config #1 plumbing: 2-process plaintext FedAvg over gloo/CPU.

Checks that one FL round (local train -> all-reduce-average of weights ->
load back) leaves every rank with identical weights equal to the mean of the
per-client post-training weights.
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
    from hefl.fl.aggregate import plaintext_fedavg
    from hefl.fl.client import LocalClient
    from hefl.parallel.dist import init_distributed

    init_distributed(backend="gloo")
    cfg = preset("config1")
    cfg.fl.samples_per_client = 64
    cfg.train.local_epochs = 1
    cfg.train.batch_size = 32
    client = LocalClient(cfg, client_id=rank, device="cpu")
    stats = client.local_train()
    assert stats.steps == 2  # 64 samples / bs 32
    local = client.get_weights().clone()
    # gather everyone's local weights to verify the average independently
    gathered = [torch.zeros_like(local) for _ in range(world)]
    dist.all_gather(gathered, local)
    avg = plaintext_fedavg(client.get_weights())
    client.set_weights(avg)
    expect = torch.stack(gathered).mean(0)
    ok = torch.allclose(client.get_weights(), expect, atol=1e-6)
    # clients started from identical init but trained on different shards
    differ = not torch.allclose(gathered[0], gathered[1])
    q.put((rank, bool(ok), bool(differ)))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_two_process_plaintext_fedavg():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    from conftest import free_port
    port = free_port()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    assert all(ok for _, ok, _ in results), results
    assert all(d for _, _, d in results), "shards did not differ"


def _equiv_worker(rank, world, port, q):
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    from hefl.config import preset
    from hefl.fl.round import FLRunner
    from hefl.parallel.dist import init_distributed

    init_distributed(backend="gloo")
    cfg = preset("config1")
    cfg.fl.n_clients = world
    cfg.fl.samples_per_client = 64
    runner = FLRunner(cfg, device="cpu", rank=rank)
    runner.run_round(epochs=1)
    q.put((rank, runner.client.get_weights()))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_distributed_matches_sequential():
    """The 1-GPU=1-client distributed round must produce the SAME global
    model as the single-process sequential simulation (plaintext: both are
    deterministic; this is the correctness anchor for the driver's
    multi-GPU scaling bench, which only changes the transport)."""
    from hefl.config import preset
    from hefl.fl.sequential import SequentialFL

    cfg = preset("config1")
    cfg.fl.n_clients = 2
    cfg.fl.samples_per_client = 64
    seq = SequentialFL(cfg, device="cpu")
    seq.run_round(epochs=1)
    from hefl.fl.weights import flat_params
    seq_vec = flat_params(seq.global_model)

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    from conftest import free_port
    port = free_port()
    procs = [ctx.Process(target=_equiv_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {r: v for r, v in (q.get(timeout=240) for _ in range(2))}
    for p in procs:
        p.join(timeout=60)
    assert torch.allclose(results[0], results[1], atol=0), \
        "ranks diverged after FedAvg"
    assert torch.allclose(results[0], seq_vec, atol=1e-5), \
        (results[0] - seq_vec).abs().max()
