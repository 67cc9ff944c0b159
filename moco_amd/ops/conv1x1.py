"""Standalone 1x1/s1 conv on the MFMA NT-GEMM — legacy/env-gated.

The PRODUCTION path is ops/block.py (the whole Bottleneck as one fused
Function: same GEMM kernels plus BN prologue/epilogue fusion, MIOpen
wgrad).  This standalone wrapper survives for A/B probing of the raw GEMM
(tools/conv1x1_probe.py) on the generic per-op route; enable with
MOCO_AMD_CUSTOM_CONV1X1=1.
"""

import os

import torch

from . import ext, force_fallback, has_ext


def custom_conv1x1_enabled() -> bool:
    return os.environ.get("MOCO_AMD_CUSTOM_CONV1X1", "0") == "1"


class _Conv1x1S1(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight):
        # x: [N, Ci, H, W] channels_last bf16; weight: [Co, Ci, 1, 1] fp32
        N, Ci, H, W = x.shape
        Co = weight.shape[0]
        w2d = weight.detach().view(Co, Ci).to(torch.bfloat16).contiguous()
        # channels_last [N,Ci,H,W] stores [N][H][W][Ci]: the permuted view
        # IS the dense [M, Ci] buffer (reshape is stride-free)
        x2d = x.permute(0, 2, 3, 1).reshape(N * H * W, Ci)
        out2d = ext().gemm_nt_bf16(x2d, w2d)
        ctx.save_for_backward(x2d, w2d)
        ctx.shape = (N, Ci, H, W, Co)
        out = out2d.view(N, H, W, Co).permute(0, 3, 1, 2)
        return out.contiguous(memory_format=torch.channels_last)

    @staticmethod
    def backward(ctx, g):
        x2d, w2d = ctx.saved_tensors
        N, Ci, H, W, Co = ctx.shape
        g2d = g.contiguous(memory_format=torch.channels_last) \
            .permute(0, 2, 3, 1).reshape(N * H * W, Co).to(torch.bfloat16)
        # dgrad: dx[M, Ci] = g[M, Co] @ W[Co, Ci]  (NT against W^T)
        wt = w2d.t().contiguous()
        dx2d = ext().gemm_nt_bf16(g2d, wt)
        dx = dx2d.view(N, H, W, Ci).permute(0, 3, 1, 2) \
            .contiguous(memory_format=torch.channels_last)
        # wgrad: dw[Co, Ci] = g^T @ x  (hipBLASLt TN GEMM)
        dw = torch.matmul(g2d.t().float(), x2d.float()) \
            .view(Co, Ci, 1, 1)
        return dx, dw


def conv1x1(x: torch.Tensor, conv: torch.nn.Conv2d) -> torch.Tensor:
    """Dispatch: custom MFMA path when applicable, else the module (MIOpen)."""
    w = conv.weight
    if (custom_conv1x1_enabled() and x.is_cuda and not force_fallback()
            and has_ext() and x.dtype == torch.bfloat16
            and conv.stride == (1, 1) and conv.kernel_size == (1, 1)
            and conv.bias is None
            and x.is_contiguous(memory_format=torch.channels_last)
            and x.shape[1] % 32 == 0 and w.shape[0] % 16 == 0):
        return _Conv1x1S1.apply(x, w)
    return conv(x)
