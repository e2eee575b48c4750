"""Config / data / model structure tests (CPU)."""
import pytest
import torch

from hefl.config import HEConfig, preset
from hefl.data.shard import shard_indices
from hefl.data.synthetic import ClientLoader, SyntheticMedicalImages
from hefl.models import CNN2, LeNet5, RefCNN6, build_model


def test_presets():
    for name in ["config1", "config2", "config3", "config4", "config5", "reference"]:
        cfg = preset(name)
        assert cfg.he.m & (cfg.he.m - 1) == 0
    assert preset("config2").he.m == 2 ** 13
    assert preset("config3").he.m == 2 ** 14
    assert preset("config5").he.m == 2 ** 15


def test_he_config_validation():
    with pytest.raises(ValueError):
        HEConfig(m=1000)
    with pytest.raises(ValueError):
        HEConfig(q_bits=(62,))  # violates lazy-allreduce bound


def test_shard_contiguous_equal():
    # reference semantics: ratio = len // n, shard i = [i*ratio, (i+1)*ratio)
    idx0 = shard_indices(1601, 0, 2)
    idx1 = shard_indices(1601, 1, 2)
    assert idx0.numel() == idx1.numel() == 800
    assert idx0[0] == 0 and idx0[-1] == 799
    assert idx1[0] == 800 and idx1[-1] == 1599  # tail sample 1600 dropped
    assert not set(idx0.tolist()) & set(idx1.tolist())


def test_synthetic_deterministic():
    ds1 = SyntheticMedicalImages(64, (28, 28, 1), 10, seed=7)
    ds2 = SyntheticMedicalImages(64, (28, 28, 1), 10, seed=7)
    idx = torch.arange(8)
    x1, y1 = ds1.batch(idx)
    x2, y2 = ds2.batch(idx)
    assert torch.equal(x1, x2) and torch.equal(y1, y2)
    assert x1.shape == (8, 28, 28, 1)
    assert float(x1.min()) >= 0 and float(x1.max()) <= 1


def test_loader_epochs():
    ds = SyntheticMedicalImages(100, (28, 28, 1), 10, seed=1)
    loader = ClientLoader(ds, torch.arange(70), batch_size=32, seed=3)
    batches = list(loader)
    assert len(batches) == 3  # 32 + 32 + 6
    assert sum(b[1].numel() for b in batches) == 70


def test_refcnn6_param_count():
    m = RefCNN6((256, 256, 3), 2)
    # Reference: 222,722 params (model factory row)
    assert m.n_params() == 222722
    assert m.feat_dim == 512


def test_model_shapes():
    m = CNN2((28, 28, 1), 10)
    x = torch.rand(4, 28, 28, 1)
    out = m(x)
    assert out.shape == (4, 10)

    m = LeNet5((32, 32, 3), 10)
    out = m(torch.rand(2, 32, 32, 3))
    assert out.shape == (2, 10)

    m = build_model(preset("config4").model)
    out = m(torch.rand(1, 224, 224, 1))
    assert out.shape == (1, 2)


def test_model_deterministic_init():
    a = CNN2(seed=5)
    b = CNN2(seed=5)
    for pa, pb in zip(a.parameters(), b.parameters()):
        assert torch.equal(pa, pb)


def test_data_pipeline_parity():
    from hefl.data.pipeline import get_test_data, get_train_data

    ds = SyntheticMedicalImages(200, (28, 28, 1), 2, seed=3)
    train, val = get_train_data(ds, client=0, n_clients=2, batch_size=32,
                                val_frac=0.1)
    # shard = 100 samples -> 90 train / 10 val (reference 10% split)
    assert sum(y.numel() for _, y in train) == 90
    assert sum(y.numel() for _, y in val) == 10
    test = get_test_data(ds, batch_size=64)
    xs = [x for x, _ in test]
    assert sum(x.shape[0] for x in xs) == 200
    # augmentation hook produces valid-range images
    x0, _ = next(iter(train))
    assert 0.0 <= x0.min() and x0.max() <= 1.0


def test_key_file_workflow(tmp_path):
    from hefl.fl.keys import gen_pk, gen_rekey, get_pk, get_sk

    d = str(tmp_path)
    HE = gen_pk(s=128, m=64, directory=d, scale_bits=30, q_bits=(50, 30),
                seed=4)
    ct = HE.encryptFrac(2.5)
    pub = get_pk(d)      # aggregator: public material only
    assert pub._sk is None
    s = (ct + ct) * 0.5  # homomorphic ops need no key at all
    sk = get_sk(d)       # key-holder decrypts
    ct._pyfhel = sk
    s._pyfhel = sk
    assert abs(sk.decryptFrac(s) - 2.5) < 1e-3
    gen_rekey(sk)        # reference's dead gen_rekey, functional here
    assert sk._keys is not None


def test_metrics_match_sklearn():
    # the reference computes precision/recall/f1 with sklearn
    # average='weighted' + accuracy_score (notebook cell 3)
    import numpy as np
    from sklearn.metrics import (accuracy_score, f1_score, precision_score,
                                 recall_score)

    from hefl.fl.metrics import classification_metrics

    rng = np.random.default_rng(0)
    y_true = torch.from_numpy(rng.integers(0, 4, 300))
    y_pred = torch.from_numpy(rng.integers(0, 4, 300))
    m = classification_metrics(y_true, y_pred, 4)
    assert abs(m["accuracy"] - accuracy_score(y_true, y_pred)) < 1e-9
    assert abs(m["precision"] - precision_score(y_true, y_pred,
                                                average="weighted",
                                                zero_division=0)) < 1e-9
    assert abs(m["recall"] - recall_score(y_true, y_pred, average="weighted",
                                          zero_division=0)) < 1e-9
    assert abs(m["f1"] - f1_score(y_true, y_pred, average="weighted",
                                  zero_division=0)) < 1e-9
