"""CLI driver: `python -m hefl --preset config2 --rounds 1`.

The executable equivalent of the reference's notebook cells 0-5: run the FL
experiment, print per-phase timings and the final metrics table
(precision/recall/F1/accuracy + wall time — notebook cells 4-5).

Single-process mode simulates all clients sequentially on one device
(reference execution model). Under torchrun (WORLD_SIZE > 1) each rank is one
client over RCCL/gloo.
"""
from __future__ import annotations

import argparse
import json
import time

import torch


def main():
    ap = argparse.ArgumentParser(prog="hefl")
    ap.add_argument("--preset", default="config2",
                    help="config1..config5 | reference (see hefl/config.py)")
    ap.add_argument("--rounds", type=int, default=1,
                    help="FL rounds (reference runs exactly 1, notebook cell 3)")
    ap.add_argument("--epochs", type=int, default=None,
                    help="local epochs per round (reference: 10)")
    ap.add_argument("--clients", type=int, default=None)
    ap.add_argument("--device", default=None)
    ap.add_argument("--plaintext", action="store_true",
                    help="disable HE (plaintext FedAvg)")
    ap.add_argument("--augment", choices=["none", "hflip", "full"],
                    default=None,
                    help="training augmentation ('full' = the reference's "
                         "shear 0.2 / zoom 0.2 / h-flip set)")
    ap.add_argument("--callbacks", action="store_true",
                    help="enable EarlyStopping/ReduceLROnPlateau per client")
    ap.add_argument("--checkpoint", default=None, metavar="PATH",
                    help="save round state (model+optimizer+public HE "
                         "material; sk to PATH.private) after every round "
                         "and RESUME from PATH if it exists — the round-"
                         "granularity recovery the reference gets only "
                         "implicitly from its pickle files (SURVEY.md §5)")
    ap.add_argument("--json", action="store_true", help="JSON line output")
    args = ap.parse_args()

    from hefl.config import preset
    from hefl.parallel.dist import get_rank, get_world_size, init_distributed

    cfg = preset(args.preset)
    if args.clients:
        cfg.fl.n_clients = args.clients
    if args.plaintext:
        cfg.fl.encrypted = False
    if args.augment is not None:
        cfg.fl.augment = args.augment
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")

    init_distributed()
    world = get_world_size()

    t_all = time.time()
    if world > 1:
        from hefl.fl.round import FLRunner
        cfg.fl.n_clients = world
        runner = FLRunner(cfg, device=device, verbose=True)
        for r in range(args.rounds):
            res = runner.run_round(epochs=args.epochs)
            if get_rank() == 0:
                print(f"round {r}: loss={res.train.train_loss:.4f} "
                      f"acc={res.train.train_acc:.4f} "
                      f"round_s={res.round_seconds:.2f} "
                      f"phases={ {k: round(v, 3) for k, v in res.phase_seconds.items()} }")
        return

    from hefl.fl.sequential import SequentialFL
    fl = SequentialFL(cfg, device=device, verbose=True)
    start_round = 0
    if args.checkpoint:
        import os

        from hefl.fl.checkpoint import load_round_state, save_round_state
        if os.path.exists(args.checkpoint):
            # resume: restore the GLOBAL model + round counter (clients
            # re-derive their state from the global weights each round)
            ref_client = fl.clients[0]
            start_round, _ = load_round_state(args.checkpoint,
                                              fl.global_model,
                                              ref_client.opt)
            print(f"resumed from {args.checkpoint} at round {start_round}")
    reports = []
    for r in range(start_round, args.rounds):
        rep = fl.run_round(epochs=args.epochs, use_callbacks=args.callbacks)
        reports.append(rep)
        print(f"round {r}: metrics={rep.metrics} round_s={rep.round_seconds:.2f}")
        if args.checkpoint:
            save_round_state(args.checkpoint, fl.global_model,
                             fl.clients[0].opt, r + 1)
    total = time.time() - t_all

    if not reports:  # resumed past the requested round count
        print("nothing to do: checkpoint already at round", start_round)
        return

    last = reports[-1]
    if args.json:
        print(json.dumps({"metrics": last.metrics, "total_seconds": total,
                          "rounds": args.rounds,
                          "clients": cfg.fl.n_clients,
                          "encrypted": cfg.fl.encrypted}))
    else:
        # metrics table, reference notebook cells 4-5 shape
        print("\n=== Final metrics (aggregated model, test set) ===")
        for k in ("accuracy", "precision", "recall", "f1"):
            print(f"  {k:<10} {last.metrics[k]:.4f}")
        print(f"  {'time (s)':<10} {total:.1f}")


if __name__ == "__main__":
    main()
