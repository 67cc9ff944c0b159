"""Reference-style import paths and transform classes keep working."""

import torch


def test_moco_builder_import_path():
    from moco import builder
    from moco_amd.builder import MoCo as MoCoImpl
    assert builder.MoCo is MoCoImpl
    assert callable(builder.concat_all_gather)


def test_moco_loader_import_path():
    from moco import loader
    t = loader.TwoCropsTransform(lambda x: x + 1)
    q, k = t(torch.zeros(3, 4, 4))
    assert torch.equal(q, k) and float(q.mean()) == 1.0


def test_gaussian_blur_transform():
    from moco.loader import GaussianBlur
    torch.manual_seed(0)
    x = torch.rand(3, 16, 16)
    y = GaussianBlur((1.0, 1.0))(x)
    assert y.shape == x.shape
    assert y.var() < x.var()  # smoothing


def test_force_fallback_env(monkeypatch):
    from moco_amd import ops
    monkeypatch.setenv("MOCO_AMD_FORCE_FALLBACK", "1")
    assert ops.force_fallback()
    x = torch.randn(2, 2)
    assert not ops.use_hip(x)  # cpu tensor never uses hip
