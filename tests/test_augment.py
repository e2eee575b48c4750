"""Affine augmentation (zoom/shear/h-flip) — the analog of the reference's
ImageDataGenerator transforms (shear 0.2, zoom 0.2, horizontal_flip;
FLPyfhelin.py:80-86), applied inside the synthetic data generator."""
import torch

from hefl.config import preset
from hefl.data.synthetic import ClientLoader, SyntheticMedicalImages


def make_ds(**kw):
    return SyntheticMedicalImages(64, (16, 16, 1), 2, seed=5, **kw)


def test_no_affine_is_identity():
    ds1, ds2 = make_ds(), make_ds()
    idx = torch.arange(8)
    x1, _ = ds1.batch(idx)
    x2, _ = ds2.batch(idx, affine=(0.0, 0.0, False))
    assert torch.equal(x1, x2)


def test_affine_changes_data_deterministically():
    ds1, ds2 = make_ds(), make_ds()
    idx = torch.arange(8)
    base, _ = make_ds().batch(idx)
    a1, _ = ds1.batch(idx, affine=(0.2, 0.2, True))
    a2, _ = ds2.batch(idx, affine=(0.2, 0.2, True))
    assert torch.equal(a1, a2)           # same seed, same call count
    assert not torch.equal(a1, base)     # transform actually applied
    assert a1.min() >= 0 and a1.max() <= 1
    # a second call draws fresh per-sample transforms (counter advances)
    a3, _ = ds1.batch(idx, affine=(0.2, 0.2, True))
    assert not torch.equal(a1, a3)


def test_hflip_only_affine_is_exact_mirror():
    ds = make_ds()
    tmpl = ds.templates[torch.zeros(16, dtype=torch.long)]
    out = ds._affine_sample(tmpl, 0.0, 0.0, True)
    # each sample is either the template or its exact horizontal mirror
    # (align_corners grid flip maps exactly onto reversed pixels)
    flipped = tmpl.flip(2)
    for i in range(out.shape[0]):
        same = torch.allclose(out[i], tmpl[i], atol=1e-6)
        mirror = torch.allclose(out[i], flipped[i], atol=1e-6)
        assert same or mirror, i
    # and both outcomes occur over 16 draws (p(miss) = 2^-16)
    n_mirror = sum(torch.allclose(out[i], flipped[i], atol=1e-6)
                   for i in range(16))
    assert 0 < n_mirror < 16


def test_reference_preset_enables_full_augment():
    cfg = preset("reference")
    assert cfg.fl.augment == "full"
    assert preset("config2").fl.augment == "none"


def test_loader_affine_applies_to_train_only():
    from hefl.config import FLConfig, ModelConfig, RunConfig
    from hefl.fl.client import LocalClient
    cfg = RunConfig(model=ModelConfig("cnn2", (12, 12, 1), 2),
                    fl=FLConfig(n_clients=1, samples_per_client=32,
                                val_samples_per_client=8, encrypted=False,
                                augment="full"))
    c = LocalClient(cfg, client_id=0)
    assert c.loader.affine == (0.2, 0.2, True)
    assert c.val_loader.affine is None
    # training still runs and learns through the augmented generator
    s = c.local_train(epochs=2)
    assert s.steps > 0 and s.train_loss > 0


def test_file_image_dataset(tmp_path):
    """folder/<class>/*.npy dataset: prep_df indexing, uint8 rescale,
    loader integration and in-loader affine (the reference's
    flow_from_dataframe workflow, FLPyfhelin.py:57-114, on .npy files)."""
    import numpy as np
    from hefl.data import FileImageDataset, get_test_data, get_train_data
    rng = np.random.default_rng(3)
    for cls in ("covid", "normal"):
        d = tmp_path / cls
        d.mkdir()
        for i in range(6):
            arr = (rng.random((12, 12, 1)) * 255).astype(np.uint8)
            np.save(d / f"img{i}.npy", arr)
    ds = FileImageDataset(str(tmp_path), seed=4)
    assert ds.n_samples == 12 and ds.n_classes == 2
    assert (ds.H, ds.W, ds.C) == (12, 12, 1)
    x, y = ds.batch(torch.arange(4))
    assert x.shape == (4, 12, 12, 1) and x.max() <= 1.0 and x.min() >= 0.0
    assert y.tolist() == [0, 0, 0, 0]  # sorted classes: covid first
    # same train/val pipeline functions as the synthetic datasets
    train, val = get_train_data(ds, client=0, n_clients=2, batch_size=4,
                                affine=(0.2, 0.2, True))
    xb, yb = next(iter(train))
    assert xb.shape[0] == 4 and xb.min() >= 0
    test = get_test_data(ds, batch_size=5)
    xt, _ = next(iter(test))
    assert xt.shape == (5, 12, 12, 1)
    # affine is deterministic per (seed, call count): two FRESH instances
    # making the same call sequence agree (ds above already advanced its
    # augment counter through the train loader)
    ds1 = FileImageDataset(str(tmp_path), seed=4)
    ds2 = FileImageDataset(str(tmp_path), seed=4)
    a1, _ = ds1.batch(torch.arange(6, 12), affine=(0.2, 0.2, True))
    a2, _ = ds2.batch(torch.arange(6, 12), affine=(0.2, 0.2, True))
    assert torch.equal(a1, a2)
    assert not torch.equal(a1, ds1.batch(torch.arange(6, 12))[0])
