"""Fused stem max-pool (3x3, stride 2, pad 1) — NHWC bf16 HIP kernels with
an argmax-tap record for an atomic-free gather backward; falls back to
F.max_pool2d elsewhere."""

import torch
import torch.nn.functional as F

from . import ext, force_fallback, has_ext


class _MaxPool3x3s2(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y, idx = ext().maxpool_fwd(x)
        ctx.save_for_backward(idx)
        ctx.in_hw = (x.shape[2], x.shape[3])
        return y

    @staticmethod
    def backward(ctx, g):
        (idx,) = ctx.saved_tensors
        H, W = ctx.in_hw
        return ext().maxpool_bwd(
            g.contiguous(memory_format=torch.channels_last), idx, H, W)


def stem_maxpool(x: torch.Tensor) -> torch.Tensor:
    """MaxPool2d(kernel_size=3, stride=2, padding=1)."""
    if (x.is_cuda and x.dtype == torch.bfloat16 and not force_fallback()
            and has_ext() and x.shape[1] % 8 == 0
            and x.is_contiguous(memory_format=torch.channels_last)):
        return _MaxPool3x3s2.apply(x)
    return F.max_pool2d(x, kernel_size=3, stride=2, padding=1)


class _HeadAvgPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.in_hw = (x.shape[2], x.shape[3])
        return ext().avgpool_fwd(x)

    @staticmethod
    def backward(ctx, g):
        H, W = ctx.in_hw
        return ext().avgpool_bwd(
            g.contiguous(memory_format=torch.channels_last), H, W)


def head_avgpool(x: torch.Tensor) -> torch.Tensor:
    """AdaptiveAvgPool2d((1, 1)) — the ResNet global pool."""
    if (x.is_cuda and x.dtype == torch.bfloat16 and not force_fallback()
            and has_ext() and x.shape[1] % 8 == 0
            and x.is_contiguous(memory_format=torch.channels_last)):
        return _HeadAvgPool.apply(x)
    return F.adaptive_avg_pool2d(x, (1, 1))
