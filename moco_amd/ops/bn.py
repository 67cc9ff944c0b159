"""Fused BatchNorm(+residual-add)(+ReLU) — the conv-block epilogue.

GPU path (bf16 + channels_last): hand-written HIP kernels (ops/hip/bn.hip),
replacing MIOpen's spatial-BN kernels plus the separate add/ReLU elementwise
launches (together 32 ms of a 58 ms ResNet-50 step — see
profiles/r01_step_breakdown.md).  Anything else falls back to
F.batch_norm (+ add + relu), which is also the numerics oracle.
"""

from typing import Optional

import torch
import torch.nn.functional as F

from . import ext, force_fallback, has_ext


class _FusedBNAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, training,
                momentum, eps, relu, residual):
        if training:
            y, mean, rstd = ext().bn_fwd_train(
                x, residual, gamma, beta, running_mean, running_var, eps,
                momentum, relu)
        else:
            mean = running_mean
            rstd = torch.rsqrt(running_var + eps)
            y = ext().bn_fwd_eval(x, residual, gamma, beta, mean, rstd, relu)
        ctx.save_for_backward(x, y, gamma, mean, rstd)
        ctx.relu = relu
        ctx.training = training
        ctx.has_res = residual is not None
        return y

    @staticmethod
    def backward(ctx, g):
        x, y, gamma, mean, rstd = ctx.saved_tensors
        outs = ext().bn_bwd(g.contiguous(memory_format=torch.channels_last),
                            y, x, mean, rstd, gamma, ctx.relu, ctx.training,
                            ctx.has_res)
        dx, dgamma, dbeta = outs[0], outs[1], outs[2]
        dres = outs[3] if ctx.has_res else None
        return (dx, dgamma, dbeta, None, None, None, None, None, None, dres)


def _bn_c_supported(C: int) -> bool:
    """kernel geometry: Q = C/8 channel octets must divide a 256-thread
    block (C in {64, 128, 256, 512, 1024, 2048})."""
    return C % 8 == 0 and 1 <= C // 8 <= 256 and 256 % (C // 8) == 0


def bn_act(x: torch.Tensor, bn: torch.nn.BatchNorm2d, relu: bool = True,
           residual: Optional[torch.Tensor] = None) -> torch.Tensor:
    """y = relu?(bn(x) + residual?) with nn.BatchNorm2d semantics
    (biased batch var for normalization, unbiased for the running stats)."""
    training = bn.training
    if training and bn.num_batches_tracked is not None \
            and not getattr(bn, "_nbt_managed", False):
        bn.num_batches_tracked += 1

    use_fused = (x.is_cuda and x.dtype == torch.bfloat16
                 and not force_fallback() and has_ext()
                 and _bn_c_supported(x.shape[1])
                 and x.is_contiguous(memory_format=torch.channels_last)
                 and (residual is None or
                      (residual.dtype == torch.bfloat16 and
                       residual.is_contiguous(
                           memory_format=torch.channels_last))))
    if use_fused:
        return _FusedBNAct.apply(x, bn.weight, bn.bias, bn.running_mean,
                                 bn.running_var, training, bn.momentum,
                                 bn.eps, relu, residual)

    out = F.batch_norm(x, bn.running_mean, bn.running_var, bn.weight,
                       bn.bias, training, bn.momentum, bn.eps)
    if residual is not None:
        out = out + residual
    return F.relu(out) if relu else out
