"""hefl flagship benchmark: encrypted federated CNN training on MI355X.

Measures BASELINE.json's headline metric — FL rounds/sec (and
samples/sec/client) for CKKS-encrypted FedAvg of a CNN on 28x28 synthetic
data — config #2, with N GPUs = N federated clients (weak scaling: each
client always trains its own fixed-size shard).

One "step" = one full FL round exactly as the reference's notebook cell 3
defines it: 10 local epochs over the client's 720-sample shard at batch
size 32 (FLPyfhelin.py:31-33,179-198), then CKKS encrypt -> ciphertext
all-reduce over xGMI -> 1/n scale + rescale -> decrypt -> load weights.
The reference measured 6583.6 s for such a round =
1.52e-4 rounds/sec.

Launch (driver contract):
  python bench.py                      # 1 GPU, quick defaults
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N --steps K --warmup W
"""
from __future__ import annotations

import argparse
import json
import sys
import time

import torch

BASELINE_ROUNDS_PER_SEC = 1.52e-4  # reference round: 6583.6 s


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--preset", type=str, default="config2")
    ap.add_argument("--local-epochs", type=int, default=None,
                    help="override local epochs per round (reference: 10)")
    args = ap.parse_args()

    from hefl.config import preset
    from hefl.fl.round import FLRunner
    from hefl.parallel.dist import barrier, get_rank, get_world_size, init_distributed

    has_gpu = torch.cuda.is_available()
    local_rank = init_distributed()
    world = get_world_size()
    rank = get_rank()
    device = f"cuda:{local_rank}" if has_gpu else "cpu"
    if has_gpu:
        torch.cuda.set_device(local_rank)
        import hefl
        hefl.load_extension()  # fail loudly if the HIP extension is missing

    cfg = preset(args.preset)
    cfg.fl.n_clients = max(world, 1)
    epochs = args.local_epochs
    if epochs is None:
        epochs = cfg.train.local_epochs  # reference default: 10

    runner = FLRunner(cfg, device=device, rank=rank)

    def sync():
        if has_gpu:
            torch.cuda.synchronize()
        barrier()

    for _ in range(args.warmup):
        runner.run_round(epochs=epochs)

    sync()
    t0 = time.perf_counter()
    results = [runner.run_round(epochs=epochs) for _ in range(args.steps)]
    sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    t = torch.tensor([elapsed], dtype=torch.float64)
    if world > 1:
        import torch.distributed as dist
        if has_gpu:
            tg = t.to(device)
            dist.all_reduce(tg, op=dist.ReduceOp.MAX)
            t = tg.cpu()
        else:
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    rounds_per_sec = args.steps / elapsed
    samples_per_round = sum(r.train.samples for r in results) / max(args.steps, 1)
    samples_per_sec_per_client = samples_per_round * rounds_per_sec

    if rank == 0:
        out = {
            "metric": "fl_rounds_per_sec",
            "value": rounds_per_sec,
            "unit": "rounds/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": rounds_per_sec / BASELINE_ROUNDS_PER_SEC,
            "dtype": "bf16" if has_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": cfg.model.name,
                "image": "x".join(map(str, cfg.model.in_shape)),
                "global_batch": cfg.train.batch_size * max(world, 1),
                "local_epochs": epochs,
                "samples_per_client": cfg.fl.samples_per_client,
                "samples_per_sec_per_client": samples_per_sec_per_client,
                "augment": cfg.fl.augment,
                "he": {"scheme": "CKKS", "m": cfg.he.m,
                       "q_bits": list(cfg.he.q_bits),
                       "scale_bits": cfg.he.scale_bits,
                       "encrypted": cfg.fl.encrypted},
                "parallelism": f"fl-dp{max(world, 1)} (1 GPU = 1 client)",
                "device": device.split(":")[0],
            },
        }
        print(json.dumps(out))
        sys.stdout.flush()


if __name__ == "__main__":
    main()
