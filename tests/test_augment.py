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
