"""Round-2 correctness fixes: key agreement via broadcast, sk hygiene,
disjoint test data, scale checks, validation split, structural N=1/N>1
identity of the aggregation op sequence.
"""
import multiprocessing as mp
import os
import pickle

import pytest
import torch

from hefl.config import FLConfig, HEConfig, ModelConfig, RunConfig, preset
from hefl.he.ckks import CKKSContext, CtxtTensor
from hefl.he.pyfhel_compat import Pyfhel


# ---------------------------------------------------------------------------
# Key agreement: with he.seed=None every rank keygens from OS entropy; the
# rank-0 broadcast in SecureAggregator must still give one shared keypair.
# ---------------------------------------------------------------------------

def _worker_unseeded(rank, world, port, q):
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    from hefl.fl.secure import SecureAggregator
    from hefl.parallel.dist import init_distributed

    init_distributed(backend="gloo")
    cfg = HEConfig(m=128, scale_bits=30, q_bits=(50, 30), seed=None)  # OS entropy
    agg = SecureAggregator(CKKSContext(cfg), rank=rank)
    vec = torch.randn(300, generator=torch.Generator().manual_seed(7 + rank))
    out = agg.fedavg(vec, n_clients=world)
    expect = torch.stack(
        [torch.randn(300, generator=torch.Generator().manual_seed(7 + r))
         for r in range(world)]).mean(0)
    q.put((rank, (out - expect).abs().max().item()))
    dist.destroy_process_group()


def _worker_bucketed(rank, world, port, q):
    """Tiny bucket size forces the multi-slab pipelined path; result must
    equal the plain mean."""
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    from hefl.fl.secure import SecureAggregator
    from hefl.parallel.dist import init_distributed

    init_distributed(backend="gloo")
    cfg = HEConfig(m=128, scale_bits=30, q_bits=(50, 30), seed=11)
    # bucket of one ciphertext -> vector of 300 values / 64 slots = 5 slabs
    agg = SecureAggregator(CKKSContext(cfg), rank=rank, bucket_bytes=1)
    vec = torch.randn(300, generator=torch.Generator().manual_seed(21 + rank))
    out = agg.fedavg(vec, n_clients=world)
    expect = torch.stack(
        [torch.randn(300, generator=torch.Generator().manual_seed(21 + r))
         for r in range(world)]).mean(0)
    q.put((rank, (out - expect).abs().max().item()))
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
def test_unseeded_keygen_agrees_via_broadcast():
    from conftest import free_port
    for rank, err in _run(_worker_unseeded, 2, free_port()):
        assert err < 1e-3, (rank, err)


@pytest.mark.timeout(300)
def test_bucketed_allreduce_matches_mean():
    from conftest import free_port
    for rank, err in _run(_worker_bucketed, 2, free_port()):
        assert err < 1e-3, (rank, err)


# ---------------------------------------------------------------------------
# N=1 / N>1 structural identity: the divide (ct x 1/n + rescale) runs even
# for a single client, so the op sequence does not depend on world size.
# ---------------------------------------------------------------------------

def test_single_client_fedavg_runs_divide():
    from hefl.fl.secure import SecureAggregator
    ctx = CKKSContext(HEConfig(m=128, scale_bits=30, q_bits=(50, 30), seed=3))
    agg = SecureAggregator(ctx, rank=0)
    vec = torch.randn(200, generator=torch.Generator().manual_seed(5))
    ct = agg.fedavg_ct(vec, n_clients=1)
    # rescale happened: level dropped from L to L-1 exactly as in the N=8 path
    assert ct.level == ctx.L - 1
    out = agg.decrypt(ct)
    assert (out - vec).abs().max().item() < 1e-3


# ---------------------------------------------------------------------------
# Secret-key hygiene
# ---------------------------------------------------------------------------

def test_publickey_pickle_cannot_decrypt(tmp_path):
    from hefl.fl.keys import gen_pk, get_pk, get_sk
    d = str(tmp_path)
    he = gen_pk(s=128, m=64, directory=d, scale_bits=30, q_bits=(50, 30),
                seed=9)
    ct = he.encryptFrac(3.25)
    pub = get_pk(d)
    assert pub._sk is None
    with pytest.raises(ValueError):
        pub.decryptFrac(ct)
    # and nothing inside the public artifact reconstructs a decryptor
    with open(os.path.join(d, "publickey.pickle"), "rb") as f:
        blob = pickle.load(f)
    assert "sk" not in blob and blob["HE"]._sk is None
    # the private artifact still decrypts
    assert abs(get_sk(d).decryptFrac(ct) - 3.25) < 1e-4


def test_round_checkpoint_excludes_sk(tmp_path):
    from hefl.fl.checkpoint import load_round_state, save_round_state
    from hefl.models import build_model
    from hefl.ops.adam import FusedAdam
    he = Pyfhel()
    he.contextGen(m=64, scale_bits=30, q_bits=(50, 30), seed=4)
    he.keyGen()
    ct = he.encryptFrac(1.5)
    model = build_model(ModelConfig("cnn2", (28, 28, 1), 10))
    opt = FusedAdam(model.parameters(), lr=1e-3)
    path = str(tmp_path / "round.pt")
    save_round_state(path, model, opt, 3, he=he)
    # the round file holds only public HE material
    state = torch.load(path, weights_only=False)
    assert "secret_key" not in state["he"]
    # a reader of the round file alone cannot decrypt
    he_pub = Pyfhel()
    he_pub.from_bytes_context(state["he"]["context"])
    he_pub.from_bytes_publicKey(state["he"]["public_key"])
    with pytest.raises(ValueError):
        he_pub.decryptFrac(ct)
    # full resume (round file + private artifact) restores decryption
    he2 = Pyfhel()
    model2 = build_model(ModelConfig("cnn2", (28, 28, 1), 10), seed=1)
    opt2 = FusedAdam(model2.parameters(), lr=1e-3)
    rnd, _ = load_round_state(path, model2, opt2, he=he2)
    assert rnd == 3
    assert abs(he2.decryptFrac(ct) - 1.5) < 1e-4


def test_pyctxt_add_zero_returns_clone():
    he = Pyfhel()
    he.contextGen(m=64, scale_bits=30, q_bits=(50, 30), seed=2)
    he.keyGen()
    ct = he.encryptFrac(2.0)
    acc = 0 + ct            # reference-style accumulator seed (ndarray of 0s)
    assert acc is not ct
    acc._ct.data.add_(1)    # mutate the accumulator in place
    assert abs(he.decryptFrac(ct) - 2.0) < 1e-4  # original ct unharmed


# ---------------------------------------------------------------------------
# CKKS guards
# ---------------------------------------------------------------------------

def test_add_rejects_scale_mismatch():
    ctx = CKKSContext(HEConfig(m=64, scale_bits=30, q_bits=(50, 30, 30),
                               seed=6))
    kp = ctx.keygen()
    import numpy as np
    a = ctx.encrypt(ctx.encode(np.ones(4)), kp.pk)
    b = ctx.encrypt(ctx.encode(np.ones(4)), kp.pk)
    bad = ctx.mul_scalar(b, 0.5)  # scale is now Delta^2
    with pytest.raises(ValueError):
        ctx.add(a, bad)


def test_sequential_rejects_more_than_8_clients():
    from hefl.fl.sequential import SequentialFL
    cfg = preset("config2")
    cfg.he.m = 64
    cfg.he.seed = 1
    cfg.fl.n_clients = 9
    cfg.fl.samples_per_client = 8
    cfg.fl.val_samples_per_client = 0
    fl = SequentialFL(cfg, device="cpu")
    with pytest.raises(AssertionError):
        fl.run_round(epochs=0)


# ---------------------------------------------------------------------------
# Disjoint test data (same task, fresh draws)
# ---------------------------------------------------------------------------

def test_heldout_test_set_shares_templates_not_labels():
    from hefl.data.synthetic import SyntheticMedicalImages
    from hefl.fl.sequential import TEST_SEED_OFFSET
    train = SyntheticMedicalImages(400, (8, 8, 1), 2, seed=1234)
    test = SyntheticMedicalImages(400, (8, 8, 1), 2,
                                  seed=1234 + TEST_SEED_OFFSET,
                                  template_seed=1234)
    # same classification task ...
    assert torch.equal(train.templates, test.templates)
    # ... but not the same label stream
    assert not torch.equal(train.labels, test.labels)


# ---------------------------------------------------------------------------
# Validation split drives the callbacks
# ---------------------------------------------------------------------------

def test_client_trains_on_train_subset_only():
    from hefl.fl.client import LocalClient
    cfg = RunConfig(model=ModelConfig("cnn2", (8, 8, 1), 2),
                    fl=FLConfig(n_clients=2, samples_per_client=40,
                                val_samples_per_client=8, encrypted=False))
    c = LocalClient(cfg, client_id=0)
    assert c.loader.indices.numel() == 40
    assert c.val_loader.indices.numel() == 8
    # the val indices are the TRAILING slice of this client's shard
    # (reference validation_split semantics) and disjoint from training
    assert set(c.loader.indices.tolist()).isdisjoint(
        set(c.val_loader.indices.tolist()))
    assert c.val_loader.indices.min().item() == 40


def test_callbacks_react_to_val_metrics():
    from hefl.fl.callbacks import EarlyStopping, ReduceLROnPlateau
    from hefl.fl.client import LocalClient
    cfg = RunConfig(model=ModelConfig("cnn2", (12, 12, 1), 2),
                    fl=FLConfig(n_clients=1, samples_per_client=64,
                                val_samples_per_client=16, encrypted=False))
    c = LocalClient(cfg, client_id=0)
    seen = []

    class Spy(EarlyStopping):
        def on_epoch_end(self, epoch, logs):
            seen.append(dict(logs))
            super().on_epoch_end(epoch, logs)

    es = Spy(c.model, monitor="val_loss", patience=1, restore_best=True)
    rl = ReduceLROnPlateau(c.opt, monitor="val_loss", factor=0.3, patience=1)
    c.local_train(epochs=3, callbacks=[es, rl])
    assert seen and all("val_loss" in lg and "val_accuracy" in lg
                        for lg in seen)
    # ES tracked the val metric (its best is a real val_loss it saw)
    assert es.best == min(lg["val_loss"] for lg in seen)


def _worker_four(rank, world, port, q):
    os.environ.update({
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    import torch.distributed as dist
    from hefl.fl.secure import SecureAggregator
    from hefl.parallel.dist import init_distributed

    init_distributed(backend="gloo")
    # unseeded keygen + multi-slab pipeline at world 4 (broadcast + bucket
    # paths together, above the 2-rank smoke level)
    cfg = HEConfig(m=128, scale_bits=30, q_bits=(50, 30), seed=None)
    agg = SecureAggregator(CKKSContext(cfg), rank=rank, bucket_bytes=4096)
    vec = torch.randn(500, generator=torch.Generator().manual_seed(100 + rank))
    out = agg.fedavg(vec, n_clients=world)
    expect = torch.stack(
        [torch.randn(500, generator=torch.Generator().manual_seed(100 + r))
         for r in range(world)]).mean(0)
    q.put((rank, (out - expect).abs().max().item()))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_four_rank_unseeded_bucketed_fedavg():
    from conftest import free_port
    for rank, err in _run(_worker_four, 4, free_port()):
        assert err < 1e-3, (rank, err)


def test_round_checkpoint_custom_private_path(tmp_path):
    """save/load_round_state honor an explicit private-key artifact path
    (the decrypting party may keep sk on separate storage)."""
    from hefl.fl.checkpoint import load_round_state, save_round_state
    from hefl.models import build_model
    from hefl.ops.adam import FusedAdam
    he = Pyfhel()
    he.contextGen(m=64, scale_bits=30, q_bits=(50, 30), seed=8)
    he.keyGen()
    ct = he.encryptFrac(2.5)
    model = build_model(ModelConfig("cnn2", (28, 28, 1), 10))
    opt = FusedAdam(model.parameters(), lr=1e-3)
    rp = str(tmp_path / "round.pt")
    sp = str(tmp_path / "keys" / "sk.pt")
    os.makedirs(os.path.dirname(sp))
    save_round_state(rp, model, opt, 1, he=he, private_path=sp)
    assert os.path.exists(sp) and not os.path.exists(rp + ".private")
    he2 = Pyfhel()
    m2 = build_model(ModelConfig("cnn2", (28, 28, 1), 10), seed=2)
    o2 = FusedAdam(m2.parameters(), lr=1e-3)
    rnd, _ = load_round_state(rp, m2, o2, he=he2, private_path=sp)
    assert rnd == 1
    assert abs(he2.decryptFrac(ct) - 2.5) < 1e-4
