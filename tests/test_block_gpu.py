"""Fused Bottleneck (conv+BN fusion) vs the generic per-op path, judged
against a pure fp32 oracle: the fused path must be as accurate as the
generic bf16 path it replaces.  All @gpu."""

import copy
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


def _run(block, x, mode: str):
    """One forward+backward; mode in {fused, generic, fp32}.
    Returns (y fp32 cpu, grads dict fp32 cpu)."""
    block.zero_grad(set_to_none=True)
    xc = x.detach().clone().requires_grad_(True)
    if mode == "fp32":
        with mock.patch("moco_amd.models.resnet.fused_bottleneck",
                        return_value=None):
            y = block(xc)
    else:
        patch = (mock.patch("moco_amd.models.resnet.fused_bottleneck",
                            return_value=None)
                 if mode == "generic" else mock.patch("os.path.curdir"))
        with torch.autocast("cuda", dtype=torch.bfloat16), patch:
            y = block(xc)
    g = torch.linspace(-1, 1, y.numel(), device=y.device) \
        .view_as(y).to(y.dtype)
    y.backward(g)
    grads = {n: p.grad.detach().float().cpu()
             for n, p in block.named_parameters() if p.grad is not None}
    grads["__x__"] = xc.grad.detach().float().cpu()
    return y.detach().float().cpu(), grads


@pytest.mark.parametrize("inplanes,planes,stride,N,H", [
    (64, 64, 1, 2, 8),      # stride-1 with (stride-1) downsample, M = 128
    (256, 64, 1, 2, 8),     # layer1-interior: identity residual
    (256, 128, 2, 2, 16),   # layer2 first block: stride-2 + downsample
])
def test_fused_bottleneck_accuracy_vs_generic(inplanes, planes, stride, N, H):
    """err(fused vs fp32 oracle) must be comparable to err(generic bf16 vs
    fp32 oracle) for the output and every gradient."""
    blk = _make_block(inplanes, planes, stride, seed=31).cuda().train() \
        .to(memory_format=torch.channels_last)
    blk_gen = copy.deepcopy(blk)
    blk_f32 = copy.deepcopy(blk)
    torch.manual_seed(7)
    x32 = (torch.randn(N, inplanes, H, H) * 0.5).cuda() \
        .contiguous(memory_format=torch.channels_last)
    xb = x32.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)

    from moco_amd.ops.block import fused_bottleneck_applicable
    assert fused_bottleneck_applicable(xb, blk)

    y_f, g_f = _run(blk, xb, "fused")
    y_g, g_g = _run(blk_gen, xb, "generic")
    y_o, g_o = _run(blk_f32, x32, "fp32")

    def errs(a, o):
        return float((a - o).abs().max())

    ey_f, ey_g = errs(y_f, y_o), errs(y_g, y_o)
    assert ey_f <= 2.0 * ey_g + 1e-2, (ey_f, ey_g)
    for name in g_o:
        ef = errs(g_f[name], g_o[name])
        eg = errs(g_g[name], g_o[name])
        floor = 1e-2 * max(1.0, float(g_o[name].abs().max()))
        assert ef <= 2.0 * eg + floor, (name, ef, eg)

    # running stats from the fused stats epilogues track the oracle
    for bn_name in ["bn1", "bn2", "bn3"]:
        bm = dict(blk.named_buffers())[f"{bn_name}.running_mean"].cpu()
        om = dict(blk_f32.named_buffers())[f"{bn_name}.running_mean"].cpu()
        bv = dict(blk.named_buffers())[f"{bn_name}.running_var"].cpu()
        ov = dict(blk_f32.named_buffers())[f"{bn_name}.running_var"].cpu()
        assert torch.allclose(bm, om, atol=2e-2), bn_name
        assert torch.allclose(bv, ov, atol=2e-2, rtol=2e-2), bn_name
    nbt = dict(blk.named_buffers())["bn1.num_batches_tracked"]
    assert int(nbt) == 1


def test_fused_bottleneck_matches_fp32_torch_oracle():
    """Forward check against an independent hand-written fp32 reference
    (identity residual, no downsample)."""
    inplanes, planes, N, H = 256, 64, 2, 8
    blk = _make_block(inplanes, planes, seed=33).cuda().train() \
        .to(memory_format=torch.channels_last)
    assert blk.downsample is None
    torch.manual_seed(9)
    x = (torch.randn(N, inplanes, H, H) * 0.5).cuda()
    xb = x.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)

    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = blk(xb)
    assert y.dtype == torch.bfloat16

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


def test_fused_bottleneck_grad_strides_match_params():
    """FusedSGD/DDP require grad strides == param strides (channels_last
    conv weights)."""
    blk = _make_block(256, 64, seed=36).cuda().train() \
        .to(memory_format=torch.channels_last)
    x = torch.randn(2, 256, 8, 8).cuda().to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = blk(x)
    y.float().sum().backward()
    for n, p in blk.named_parameters():
        assert p.grad is not None, n
        assert p.grad.stride() == p.stride(), (n, p.grad.stride(), p.stride())


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


def test_fused_bottleneck_chain_handoff():
    """Two stacked identity-residual blocks: block 2's conv1-dgrad epilogue
    takes over block 1's bn3 backward-reduce (cross-block fusion).  Checked
    against the fp32 oracle like the single-block test."""
    inplanes, planes, N, H = 256, 64, 2, 8
    torch.manual_seed(41)
    b1 = _make_block(inplanes, planes, seed=41)
    b2 = _make_block(inplanes, planes, seed=42)
    assert b1.downsample is None and b2.downsample is None
    chain = torch.nn.Sequential(b1, b2).cuda().train() \
        .to(memory_format=torch.channels_last)
    object.__setattr__(chain[1], "_prev_bottleneck", chain[0])
    chain_gen = copy.deepcopy(chain)
    chain_f32 = copy.deepcopy(chain)

    torch.manual_seed(8)
    x32 = (torch.randn(N, inplanes, H, H) * 0.5).cuda() \
        .contiguous(memory_format=torch.channels_last)
    xb = x32.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)

    y_f, g_f = _run(chain, xb, "fused")
    # handoff engaged: block 2 consumed block 1's forward state
    assert chain[0]._bn3_fwd_state is None
    y_g, g_g = _run(chain_gen, xb, "generic")
    y_o, g_o = _run(chain_f32, x32, "fp32")

    ef, eg = float((y_f - y_o).abs().max()), float((y_g - y_o).abs().max())
    assert ef <= 2.0 * eg + 1e-2, (ef, eg)
    for name in g_o:
        e1 = float((g_f[name] - g_o[name]).abs().max())
        e2 = float((g_g[name] - g_o[name]).abs().max())
        floor = 1e-2 * max(1.0, float(g_o[name].abs().max()))
        assert e1 <= 2.0 * e2 + floor, (name, e1, e2)
