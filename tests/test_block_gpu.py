"""Fused Bottleneck (conv+BN fusion) vs the generic per-op path and a pure
fp32 torch oracle.  All @gpu."""

from unittest import mock

import pytest
import torch

from moco_amd.models.resnet import Bottleneck, conv1x1

pytestmark = pytest.mark.gpu


def _make_block(inplanes, planes, stride=1, seed=0):
    torch.manual_seed(seed)
    downsample = None
    if stride != 1 or inplanes != planes * Bottleneck.expansion:
        downsample = torch.nn.Sequential(
            conv1x1(inplanes, planes * Bottleneck.expansion, stride),
            torch.nn.BatchNorm2d(planes * Bottleneck.expansion))
    b = Bottleneck(inplanes, planes, stride, downsample)
    # non-trivial BN params so the apply/backward paths are exercised
    for bn in [b.bn1, b.bn2, b.bn3] + (
            [downsample[1]] if downsample is not None else []):
        torch.nn.init.uniform_(bn.weight, 0.5, 1.5)
        torch.nn.init.uniform_(bn.bias, -0.3, 0.3)
    return b


def _clone_block(b):
    import copy
    return copy.deepcopy(b)


def _run(block, x, fused: bool):
    """One forward+backward through the block; returns (y, grads dict)."""
    block.zero_grad(set_to_none=True)
    xc = x.detach().clone().requires_grad_(True)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        if fused:
            y = block(xc)
        else:
            with mock.patch("moco_amd.models.resnet.fused_bottleneck",
                            return_value=None):
                y = block(xc)
    # deterministic upstream grad
    g = torch.linspace(-1, 1, y.numel(), device=y.device) \
        .view_as(y).to(y.dtype)
    y.backward(g)
    grads = {n: p.grad.detach().float().cpu()
             for n, p in block.named_parameters() if p.grad is not None}
    grads["__x__"] = xc.grad.detach().float().cpu()
    return y.detach().float().cpu(), grads


@pytest.mark.parametrize("inplanes,planes,stride,N,H", [
    (64, 64, 1, 2, 8),      # layer1-interior shape class, M = 128
    (256, 64, 1, 2, 8),     # layer1 interior (256 in)
    (256, 128, 2, 2, 16),   # layer2 first block: stride-2 + downsample
])
def test_fused_bottleneck_matches_generic(inplanes, planes, stride, N, H):
    blk = _make_block(inplanes, planes, stride, seed=31).cuda().train()
    blk = blk.to(memory_format=torch.channels_last)
    blk_ref = _clone_block(blk)
    torch.manual_seed(7)
    x = (torch.randn(N, inplanes, H, H) * 0.5).cuda().to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)

    from moco_amd.ops.block import fused_bottleneck_applicable
    assert fused_bottleneck_applicable(x, blk)

    y_f, g_f = _run(blk, x, fused=True)
    y_r, g_r = _run(blk_ref, x, fused=False)

    assert (y_f - y_r).abs().max() < 8e-2, (y_f - y_r).abs().max()
    for name in g_r:
        a, b = g_f[name], g_r[name]
        scale = b.abs().max().clamp_min(1e-3)
        d = (a - b).abs().max() / scale
        assert d < 8e-2, (name, float(d), float(scale))

    # running stats from the fused stats epilogues match the generic path
    for bn_name in ["bn1", "bn2", "bn3"]:
        bm = dict(blk.named_buffers())[f"{bn_name}.running_mean"]
        rm = dict(blk_ref.named_buffers())[f"{bn_name}.running_mean"]
        bv = dict(blk.named_buffers())[f"{bn_name}.running_var"]
        rv = dict(blk_ref.named_buffers())[f"{bn_name}.running_var"]
        assert torch.allclose(bm.cpu(), rm.cpu(), atol=2e-2), bn_name
        assert torch.allclose(bv.cpu(), rv.cpu(), atol=2e-2,
                              rtol=2e-2), bn_name
    nbt = dict(blk.named_buffers())["bn1.num_batches_tracked"]
    assert int(nbt) == 2  # two forwards


def test_fused_bottleneck_matches_fp32_torch_oracle():
    """End-to-end check against an independent fp32 reference (not via
    bn_act), looser tolerance for the bf16 compute."""
    inplanes, planes, N, H = 64, 64, 2, 8
    blk = _make_block(inplanes, planes, seed=33).cuda().train() \
        .to(memory_format=torch.channels_last)
    torch.manual_seed(9)
    x = (torch.randn(N, inplanes, H, H) * 0.5).cuda()
    xb = x.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)

    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = blk(xb)
    assert y.dtype == torch.bfloat16

    # fp32 oracle with the same params
    def oracle(x):
        bnargs = dict(training=True)
        h = torch.conv2d(x, blk.conv1.weight.float())
        h = torch.relu(torch.nn.functional.batch_norm(
            h, None, None, blk.bn1.weight, blk.bn1.bias, **bnargs))
        h = torch.conv2d(h, blk.conv2.weight.float(), padding=1)
        h = torch.relu(torch.nn.functional.batch_norm(
            h, None, None, blk.bn2.weight, blk.bn2.bias, **bnargs))
        h = torch.conv2d(h, blk.conv3.weight.float())
        h = torch.nn.functional.batch_norm(
            h, None, None, blk.bn3.weight, blk.bn3.bias, **bnargs)
        return torch.relu(h + x)

    with torch.no_grad():
        ref = oracle(x)
    assert (y.float() - ref).abs().max() < 0.12, \
        (y.float() - ref).abs().max()


def test_fused_bottleneck_no_grad_path():
    """Key-encoder style call (no_grad): still runs fused, updates running
    stats, skips the a2 materialization."""
    blk = _make_block(64, 64, seed=35).cuda().train() \
        .to(memory_format=torch.channels_last)
    x = torch.randn(2, 64, 8, 8).cuda().to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    rm_before = blk.bn2.running_mean.clone()
    with torch.no_grad(), torch.autocast("cuda", dtype=torch.bfloat16):
        y = blk(x)
    assert y.requires_grad is False
    assert not torch.allclose(blk.bn2.running_mean, rm_before)


def test_fused_bottleneck_fallback_on_odd_shapes():
    """Non-conforming M (N*H*W % 128 != 0) must fall back, not crash."""
    blk = _make_block(64, 64, seed=37).cuda().train() \
        .to(memory_format=torch.channels_last)
    x = torch.randn(3, 64, 7, 7).cuda().to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    from moco_amd.ops.block import fused_bottleneck_applicable
    assert not fused_bottleneck_applicable(x, blk)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = blk(x)  # generic path
    assert y.shape == (3, 256, 7, 7)
