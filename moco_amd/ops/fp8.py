"""fp8 (OCP e4m3) conv path — BASELINE config-4 (SURVEY §7 stage 4).

gfx950 reaches its ~2x-over-bf16 fp8 rate only through the MX-scaled
`v_mfma_scale_f32_16x16x128_f8f6f4` instruction (the non-scaled fp8 MFMA
runs at the bf16 rate — guide §3).  This module provides per-TENSOR
scaled quantization plus a 1x1-conv forward on the fp8 GEMM kernel
(ops/hip/conv1x1.hip `gemm_nt_fp8`): operands are pre-scaled into e4m3
range, the hardware block scales are 1.0, and the epilogue multiplies the
fp32 accumulators by sa*sb.

Round-2 scope: the kernel + quantization + oracle/drift tests + perf
probe (tools/fp8_probe.py).  Full training integration (delayed scaling
so activations need no extra amax pass, fp8 dgrad) is the round-3 item —
see docs/ROADMAP_r3.md.
"""

from typing import Tuple

import torch

from . import ext, use_hip

E4M3_MAX = 448.0


def quantize_e4m3(x: torch.Tensor,
                  margin: float = 1.0) -> Tuple[torch.Tensor, float]:
    """Per-tensor scaled e4m3 quantization: returns (x8, scale) with
    x ~= x8.float() * scale."""
    amax = float(x.detach().abs().amax())
    scale = max(amax, 1e-12) / (E4M3_MAX * margin)
    x8 = (x.detach().float() / scale).clamp(-E4M3_MAX, E4M3_MAX) \
        .to(torch.float8_e4m3fn)
    return x8, scale


def gemm_nt_fp8(a8: torch.Tensor, b8t: torch.Tensor, scale_a: float,
                scale_b: float) -> torch.Tensor:
    """C_bf16[M,N] = (a8[M,K] @ b8t[N,K]^T) * scale_a * scale_b."""
    assert use_hip(a8)
    (c,) = ext().gemm_nt_fp8(a8.contiguous(), b8t.contiguous(),
                             scale_a * scale_b)
    return c


def conv1x1_fp8_fwd(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """1x1/s1 conv forward in fp8: quantize both operands per-tensor, run
    the MX-scaled GEMM, return bf16 channels_last output.  Inference /
    drift-measurement path (no autograd)."""
    n, ci, h, wd = x.shape
    x2d = x.permute(0, 2, 3, 1).reshape(n * h * wd, ci)
    w2d = w.reshape(w.shape[0], w.shape[1])
    x8, sx = quantize_e4m3(x2d)
    w8, sw = quantize_e4m3(w2d)
    out2d = gemm_nt_fp8(x8, w8, sx, sw)
    return out2d.view(n, h, wd, w.shape[0]).permute(0, 3, 1, 2)
