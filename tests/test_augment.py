import math

import torch

from moco_amd.ops.augment import (P_BR, P_CO, P_FLIP, P_GRAY, P_H, P_HU,
                                  P_ORDER, P_SA, P_SIGMA, P_W, P_X0, P_Y0,
                                  IMAGENET_MEAN, IMAGENET_STD,
                                  _augment_torch, sample_two_crop_params,
                                  two_crop_augment)


def identity_params(n, H, W, S):
    """Params that make the pipeline a pure resize+normalize."""
    p = torch.zeros(n, 12)
    p[:, P_W] = W
    p[:, P_H] = H
    p[:, P_BR] = 1.0
    p[:, P_CO] = 1.0
    p[:, P_SA] = 1.0
    return p


def test_param_sampling_shapes_and_ranges():
    for aug_plus in (False, True):
        p = sample_two_crop_params(8, 256, 256, 224, aug_plus)
        assert p.shape == (16, 12)
        assert (p[:, P_X0] >= 0).all() and (p[:, P_Y0] >= 0).all()
        assert (p[:, P_X0] + p[:, P_W] <= 256).all()
        assert (p[:, P_Y0] + p[:, P_H] <= 256).all()
        if aug_plus:
            assert (p[:, P_ORDER] == 0).all()
        else:
            assert (p[:, P_SIGMA] == 0).all()
            assert (p[:, P_HU] == 0).all()


def test_identity_crop_is_resize_normalize():
    torch.manual_seed(0)
    x = torch.rand(2, 3, 32, 32)
    p = identity_params(2, 32, 32, 32)
    out = _augment_torch(x, p, 32)
    mean = torch.tensor(IMAGENET_MEAN).view(1, 3, 1, 1)
    std = torch.tensor(IMAGENET_STD).view(1, 3, 1, 1)
    # same-size identity crop: bilinear sampling is exact
    assert torch.allclose(out, (x - mean) / std, atol=1e-5)


def test_flip():
    x = torch.rand(1, 3, 16, 16)
    p = identity_params(1, 16, 16, 16)
    p[:, P_FLIP] = 1.0
    out = _augment_torch(x, p, 16)
    ref = _augment_torch(x, identity_params(1, 16, 16, 16), 16).flip(-1)
    assert torch.allclose(out, ref, atol=1e-6)


def test_grayscale_channels_equal():
    x = torch.rand(1, 3, 16, 16)
    p = identity_params(1, 16, 16, 16)
    p[:, P_GRAY] = 1.0
    out = _augment_torch(x, p, 16)
    # undo normalize: all three channels equal before normalization
    mean = torch.tensor(IMAGENET_MEAN).view(3, 1, 1)
    std = torch.tensor(IMAGENET_STD).view(3, 1, 1)
    raw = out[0] * std + mean
    assert torch.allclose(raw[0], raw[1], atol=1e-5)
    assert torch.allclose(raw[1], raw[2], atol=1e-5)


def test_brightness_scales_pixels():
    x = torch.full((1, 3, 8, 8), 0.4)
    p = identity_params(1, 8, 8, 8)
    p[:, P_BR] = 1.5
    out = _augment_torch(x, p, 8)
    mean = torch.tensor(IMAGENET_MEAN).view(3, 1, 1)
    std = torch.tensor(IMAGENET_STD).view(3, 1, 1)
    raw = out[0] * std + mean
    assert torch.allclose(raw, torch.full_like(raw, 0.6), atol=1e-5)


def test_hue_rotation_preserves_value():
    torch.manual_seed(1)
    x = torch.rand(1, 3, 8, 8)
    p = identity_params(1, 8, 8, 8)
    p[:, P_HU] = 0.25
    out = _augment_torch(x, p, 8)
    mean = torch.tensor(IMAGENET_MEAN).view(3, 1, 1)
    std = torch.tensor(IMAGENET_STD).view(3, 1, 1)
    raw = (out[0] * std + mean).clamp(0, 1)
    # hue rotation preserves the max channel value (V in HSV)
    assert torch.allclose(raw.max(dim=0).values, x[0].max(dim=0).values,
                          atol=1e-4)


def test_blur_preserves_mean_roughly():
    torch.manual_seed(2)
    x = torch.rand(1, 3, 32, 32)
    p = identity_params(1, 32, 32, 32)
    p[:, P_SIGMA] = 1.5
    out = _augment_torch(x, p, 32)
    ref = _augment_torch(x, identity_params(1, 32, 32, 32), 32)
    # blur smooths: variance strictly drops, mean ~preserved (edge effects)
    assert out.var() < ref.var()
    assert abs(out.mean() - ref.mean()) < 0.05
    assert not torch.allclose(out, ref)


def test_two_crop_augment_api():
    torch.manual_seed(0)
    x = torch.rand(4, 3, 64, 64)
    q, k = two_crop_augment(x, out_size=32, aug_plus=True,
                            channels_last=False)
    assert q.shape == (4, 3, 32, 32) and k.shape == (4, 3, 32, 32)
    assert not torch.allclose(q, k)  # independent crops


def test_two_crop_deterministic_given_params():
    x = torch.rand(2, 3, 64, 64)
    params = sample_two_crop_params(2, 64, 64, 32, True)
    q1, k1 = two_crop_augment(x, 32, True, params=params,
                              channels_last=False)
    q2, k2 = two_crop_augment(x, 32, True, params=params,
                              channels_last=False)
    assert torch.equal(q1, q2) and torch.equal(k1, k2)


def test_uint8_source_cpu_equivalence():
    """uint8 sources must match the fp32/255 path exactly (the GPU kernel
    scales in pass 1; the CPU fallback converts up front)."""
    from moco_amd.ops.augment import two_crop_augment, sample_two_crop_params
    torch.manual_seed(3)
    u8 = (torch.rand(2, 3, 48, 48) * 255).to(torch.uint8)
    params = sample_two_crop_params(2, 48, 48, 32, True)
    q1, k1 = two_crop_augment(u8, 32, True, params=params,
                              channels_last=False)
    q2, k2 = two_crop_augment(u8.float() / 255.0, 32, True, params=params,
                              channels_last=False)
    assert q1.dtype == torch.float32
    assert torch.allclose(q1, q2, atol=1e-6)
    assert torch.allclose(k1, k2, atol=1e-6)
