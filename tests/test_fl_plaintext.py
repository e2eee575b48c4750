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
    port = 29601
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    assert all(ok for _, ok, _ in results), results
    assert all(d for _, _, d in results), "shards did not differ"
