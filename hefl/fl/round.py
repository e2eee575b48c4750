"""One federated-learning round: local training -> (encrypted) FedAvg -> load.

The reference's notebook cell 3 runs exactly this sequence once (SURVEY.md
section 3: train_clients -> export_encrypted_clients_weights ->
aggregate_encrypted_weights -> decrypt_import_weights -> evaluate), with
pickle files as transport. Here each rank IS one client (1 MI355X GPU =
1 client) and transport is torch.distributed (RCCL over xGMI / gloo on CPU).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, Optional

import torch

from ..config import RunConfig
from ..he.ckks import CKKSContext
from ..parallel.dist import get_rank, get_world_size
from .aggregate import plaintext_fedavg
from .client import LocalClient, RoundStats
from .secure import SecureAggregator


@dataclass
class FLRoundResult:
    train: RoundStats
    round_seconds: float
    phase_seconds: Dict[str, float] = field(default_factory=dict)


class FLRunner:
    """Owns one client (this rank) plus the aggregation state."""

    def __init__(self, cfg: RunConfig, device: str = "cpu",
                 rank: Optional[int] = None, verbose: bool = False):
        self.cfg = cfg
        self.rank = get_rank() if rank is None else rank
        self.device = device
        self.client = LocalClient(cfg, client_id=self.rank, device=device)
        self.agg: Optional[SecureAggregator] = None
        if cfg.fl.encrypted:
            ctx = CKKSContext(cfg.he, device=device)
            self.agg = SecureAggregator(ctx, rank=self.rank, verbose=verbose,
                                        denom_mode=cfg.fl.denom_mode,
                                        n_clients=cfg.fl.n_clients)

    def run_round(self, epochs: Optional[int] = None) -> FLRoundResult:
        t0 = time.perf_counter()
        stats = self.client.local_train(epochs)
        t1 = time.perf_counter()
        vec = self.client.get_weights()
        n = get_world_size()
        if self.agg is not None:
            new_vec = self.agg.fedavg(vec, n_clients=max(n, 1))
        else:
            new_vec = plaintext_fedavg(vec)
        self.client.set_weights(new_vec)
        t2 = time.perf_counter()
        return FLRoundResult(
            train=stats,
            round_seconds=t2 - t0,
            phase_seconds={"local_train": t1 - t0, "fedavg": t2 - t1},
        )
