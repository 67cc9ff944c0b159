"""3x3 conv on the implicit-GEMM MFMA kernel — round-2 prototype wrappers.

Not wired into the model yet (enable experiments with
MOCO_AMD_EXPERIMENTAL=1); ships with oracle tests so round-2 integration
starts from a verified kernel.

dgrad note: for stride 1, the data gradient of conv3x3(x, W) is itself a
3x3 pad-1 convolution of the output gradient with the spatially-rotated,
channel-transposed weights — so the SAME forward kernel serves dgrad:
    dx = conv3x3(g, W') with W'[ci, co, ky, kx] = W[co, ci, 2-ky, 2-kx]
(stride-2 dgrad needs gradient dilation and stays on MIOpen for now).
"""

from typing import Optional, Tuple

import torch

from . import ext


def conv3x3_fwd(x: torch.Tensor, w: torch.Tensor, stride: int = 1,
                a_scale: Optional[torch.Tensor] = None,
                a_shift: Optional[torch.Tensor] = None,
                a_relu: bool = False, want_stats: bool = False,
                neutral: Optional[torch.Tensor] = None):
    """x: [N,Ci,H,W] channels_last bf16; w: [Co,Ci,3,3] channels_last bf16.
    Returns out (+ per-channel sum/sumsq of the fp32 accumulators when
    want_stats)."""
    outs = ext().conv3x3_fwd(x, w, stride, a_scale, a_shift, a_relu,
                             want_stats, neutral)
    return outs if want_stats else outs[0]


def rotate_weights_for_dgrad(w: torch.Tensor) -> torch.Tensor:
    """W[Co,Ci,3,3] -> W'[Ci,Co,3,3] with both spatial axes flipped,
    channels_last, for the stride-1 dgrad-as-forward-conv trick."""
    return w.permute(1, 0, 2, 3).flip(2, 3) \
        .contiguous(memory_format=torch.channels_last)


def conv3x3_dgrad_s1(g: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Stride-1 data gradient via the forward kernel."""
    return conv3x3_fwd(g.contiguous(memory_format=torch.channels_last),
                       rotate_weights_for_dgrad(w), stride=1)


def bn_neutral_page(scale: torch.Tensor, shift: torch.Tensor) -> torch.Tensor:
    """The padding page for the fused BN-apply: values that the per-channel
    transform maps to exactly zero."""
    Ci = scale.numel()
    page = torch.zeros(Ci + 8, dtype=torch.bfloat16, device=scale.device)
    page[:Ci] = (-shift / scale.clamp_min(1e-12)).to(torch.bfloat16)
    return page
