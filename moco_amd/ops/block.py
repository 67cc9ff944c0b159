"""Fused Bottleneck block: conv+BN+ReLU chains as MFMA GEMMs with BN work
folded into prologues/epilogues (the round-2 conv+BN fusion).

Reference parity: the computation is exactly torchvision-style
Bottleneck.forward (conv1x1 -> bn1 -> relu -> conv3x3 -> bn2 -> relu ->
conv1x1 -> bn3 -> +identity -> relu), reference encoder injected at
main_moco.py:165.  The MI355X-native restructuring (ops/hip/conv1x1.hip):

forward (channels_last bf16, stats fp32):
  z1            = x @ W1^T                 [GEMM, bn1-STATS epilogue]
  mean1/rstd1   = finalize(stats)          [tiny kernel + running update]
  a1            = relu(bn1(z1))            [1R/1W apply pass]
  z2            = conv3x3(a1)              [MIOpen]
  mean2/rstd2   = bn_stats(z2)             [1R]
  z3, a2        = relu(bn2(z2)) @ W3^T     [GEMM, bn2-APPLY prologue,
                                            bn3-STATS epilogue, a2
                                            materialized for wgrad]
  mean3/rstd3   = finalize(stats)
  y             = relu(bn3(z3) + identity) [1R(+res)/1W apply pass]

vs the unfused path this removes one full read of z1, z2 and z3 per block
per encoder (BN stats or apply passes subsumed by the GEMMs) and ~4 kernel
launches.

backward (query encoder only):
  dz3, dgamma3, dbeta3, dres = bn_bwd(gy, y, z3)         [reduce + dx]
  gp2, dbeta2, dgamma2 = dgrad_epi(dz3, W3, a2, z2)      [GEMM; relu mask +
                                                          bn2 reduce fused]
  dz2  = bn_bwd_dx_only(gp2, z2)                          [1 pass]
  dw3  = dz3^T @ a2                                       [hipBLASLt TN]
  da1, dw2 = convolution_backward(dz2, a1, W2)            [MIOpen]
  dz1, dgamma1, dbeta1 = bn_bwd(da1, a1, z1)
  dx   = dz1 @ W1 (+ dres fused in the store tail when    [GEMM, c_add
         the identity is x itself)                         epilogue]
  dw1  = dz1^T @ x
"""

from typing import Optional

import torch
import torch.nn.functional as F

from . import ext, force_fallback, has_ext


_WS = {}  # (device, kind, N) -> pre-zeroed [256, N] fp32 workspace


def _ws_pair(device, kind, n):
    """Pre-zeroed slot-reduction workspaces, shared per (device, N): the
    consuming finalize/slot_reduce kernels re-zero them after reading, so
    no per-call fill kernels run.  Safe because every producer/consumer
    pair is stream-ordered on the compute stream."""
    key = (device, kind, n)
    ws = _WS.get(key)
    if ws is None:
        ws = (torch.zeros(256, n, dtype=torch.float32, device=device),
              torch.zeros(256, n, dtype=torch.float32, device=device))
        _WS[key] = ws
    return ws


def _as2d(t4d: torch.Tensor) -> torch.Tensor:
    """channels_last [N,C,H,W] -> the underlying dense [N*H*W, C] buffer
    (stride-free view)."""
    n, c, h, w = t4d.shape
    return t4d.permute(0, 2, 3, 1).reshape(n * h * w, c)


def _as4d(t2d: torch.Tensor, n: int, h: int, w: int) -> torch.Tensor:
    c = t2d.shape[1]
    return t2d.view(n, h, w, c).permute(0, 3, 1, 2)


def _bf16_weight(w: torch.Tensor) -> torch.Tensor:
    """[Co, Ci, 1, 1] or [Co, Ci] fp32 master -> [Co, Ci] bf16 contiguous."""
    return w.detach().reshape(w.shape[0], w.shape[1]).to(torch.bfloat16) \
        .contiguous()


def _as_weight_grad(dw2d: torch.Tensor, strides) -> torch.Tensor:
    """[Co, Ci] grad -> [Co, Ci, 1, 1] with the master conv weight's strides
    (NCHW-contiguous and channels_last have identical storage order for 1x1
    kernels, so this is a stride-only view — FusedSGD/DDP require grad
    strides == param strides)."""
    co, ci = dw2d.shape
    return dw2d.as_strided((co, ci, 1, 1), tuple(strides))


def _conv1x1_wgrad(dz2d: torch.Tensor, x2d: torch.Tensor, nhw,
                   w_strides) -> torch.Tensor:
    """dw[Co, Ci] = dz^T @ x via MIOpen's wrw igemm (convolution_backward
    with only the weight-grad requested)."""
    n, h, w = nhw
    dz4 = _as4d(dz2d, n, h, w)
    x4 = _as4d(x2d, n, h, w)
    co, ci = dz2d.shape[1], x2d.shape[1]
    # weight meta only (wrw never reads the values); channels_last strides
    wdummy = dz2d.new_empty((co, ci, 1, 1)).as_strided((co, ci, 1, 1),
                                                       (ci, 1, ci, ci))
    _, dw, _ = torch.ops.aten.convolution_backward(
        dz4, x4, wdummy, None, [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
        [False, True, False])
    return _as_weight_grad(
        dw.float().reshape(co, ci), w_strides)


class _FusedBottleneck(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, identity, w1, gamma1, beta1, w2, gamma2, beta2,
                w3, gamma3, beta3, bn_state, conv2_meta, need_grad,
                prev_bn3, prev_hand, state_sink, my_hand):
        # bn_state: (rm1, rv1, rm2, rv2, rm3, rv3, eps3tuple, mom)
        # conv2_meta: (stride, groups)
        # prev_bn3: (z3_prev, mean3_prev, rstd3_prev) of the PRECEDING
        #   bottleneck, when this block may fuse that block's bn3
        #   backward-reduce + relu mask into its conv1 dgrad epilogue
        #   (only identity-residual blocks qualify — see fused_bottleneck)
        # prev_hand: the preceding block's handoff dict — this block's
        #   backward deposits (dbeta3, dgamma3) there
        # state_sink: dict this forward fills with (y, z3, mean3, rstd3)
        #   for the NEXT block's dispatcher
        # my_hand: this block's own handoff dict — its backward checks it
        #   for a deposit from the next block
        rm1, rv1, rm2, rv2, rm3, rv3, eps, mom = bn_state
        stride, groups = conv2_meta
        N, Ci, H, W = x.shape
        M = N * H * W

        x2d = _as2d(x)
        # bf16 mirrors refreshed once per step by ops/mirrors.py; inline
        # casts only for standalone-block use (tests)
        w1b = getattr(w1, "_mirror_bf16", None)
        if w1b is None:
            w1b = _bf16_weight(w1)
        w2b = getattr(w2, "_mirror_bf16", None)
        if w2b is None:
            w2b = w2.detach().to(torch.bfloat16) \
                .contiguous(memory_format=torch.channels_last)
        w3b = getattr(w3, "_mirror_bf16", None)
        if w3b is None:
            w3b = _bf16_weight(w3)
        w1t = getattr(w1, "_mirror_bf16_t", None)
        w3t = getattr(w3, "_mirror_bf16_t", None)
        _C = ext()

        # conv1 (1x1/s1) with bn1-stats epilogue
        Co1 = w1b.shape[0]
        ws1 = _ws_pair(x.device, "stats", Co1)
        z1, s1, q1 = _C.gemm_nt_fused(x2d, w1b, None, None, False, True,
                                      False, None, ws1[0], ws1[1])
        mean1, rstd1 = _C.bn_finalize(s1, q1, M, rm1, rv1, eps[0], mom,
                                      zero_src=True)
        a1 = _C.bn_fwd_eval(z1, None, gamma1, beta1, mean1, rstd1, True)
        a1_4d = _as4d(a1, N, H, W)

        # conv2 (3x3, MIOpen)
        z2_4d = F.conv2d(a1_4d, w2b, None, (stride, stride), (1, 1), (1, 1),
                         groups)
        Ho, Wo = z2_4d.shape[2], z2_4d.shape[3]
        M2 = N * Ho * Wo
        z2 = _as2d(z2_4d)

        # bn2 stats (1 read); apply rides conv3's prologue
        mean2, rstd2 = _C.bn_stats(z2, rm2, rv2, eps[1], mom)
        scale2 = rstd2 * gamma2.detach()
        shift2 = beta2.detach() - mean2 * scale2

        # conv3 (1x1/s1): bn2-apply prologue + bn3-stats epilogue
        # (+ a2 materialized when the backward will need it for wgrad)
        Co3 = w3b.shape[0]
        ws3 = _ws_pair(x.device, "stats", Co3)
        outs = _C.gemm_nt_fused(z2, w3b, scale2, shift2, True, True,
                                need_grad, None, ws3[0], ws3[1])
        if need_grad:
            z3, s3, q3, a2 = outs
        else:
            z3, s3, q3 = outs
            a2 = None
        mean3, rstd3 = _C.bn_finalize(s3, q3, M2, rm3, rv3, eps[2], mom,
                                      zero_src=True)

        ident2d = _as2d(identity if identity is not None else x)
        y = _C.bn_fwd_eval(z3, ident2d, gamma3, beta3, mean3, rstd3, True)
        y4d = _as4d(y, N, Ho, Wo)

        if prev_bn3 is not None:
            z3p, mean3p, rstd3p = prev_bn3
        else:
            z3p = mean3p = rstd3p = x  # placeholders (unused)
        ctx.save_for_backward(x, z1, a1, z2, a2 if a2 is not None else z2,
                              z3, y, w1b, w2b, w3b, gamma1, gamma2, gamma3,
                              mean1, rstd1, mean2, rstd2, mean3, rstd3,
                              scale2, shift2, z3p, mean3p, rstd3p,
                              w1t if w1t is not None else w1b,
                              w3t if w3t is not None else w3b)
        ctx.meta = (N, H, W, Ho, Wo, stride, groups,
                    identity is not None, w1.stride(), w3.stride(),
                    prev_bn3 is not None, w1t is not None,
                    w3t is not None)
        ctx.prev_hand = prev_hand
        ctx.my_hand = my_hand
        if state_sink is not None:
            state_sink["y"] = y4d
            state_sink["z3"] = z3
            state_sink["mean3"] = mean3
            state_sink["rstd3"] = rstd3
        return y4d

    @staticmethod
    def backward(ctx, gy):
        (x, z1, a1, z2, a2, z3, y, w1b, w2b, w3b, gamma1, gamma2, gamma3,
         mean1, rstd1, mean2, rstd2, mean3, rstd3, scale2, shift2,
         z3p, mean3p, rstd3p, w1t, w3t) = ctx.saved_tensors
        (N, H, W, Ho, Wo, stride, groups, has_sep_identity, w1_strides,
         w3_strides, has_prev, has_w1t, has_w3t) = ctx.meta
        if not has_w1t:
            w1t = None
        if not has_w3t:
            w3t = None
        _C = ext()

        gy2d = _as2d(gy.contiguous(memory_format=torch.channels_last))

        # bn3 backward.  When the NEXT block's conv1-dgrad epilogue already
        # applied the relu mask and reduced dbeta3/dgamma3 (deposit in
        # my_hand), only the dx pass remains and dres IS the incoming
        # (pre-masked) grad; otherwise the full reduce + dx.
        deposit = ctx.my_hand.pop("deposit", None) \
            if ctx.my_hand is not None else None
        if deposit is not None:
            dbeta3, dgamma3 = deposit
            dz3 = _C.bn_bwd_dx_only(gy2d, z3, mean3, rstd3, gamma3,
                                    dbeta3, dgamma3)
            dres = gy2d
        else:
            dz3, dgamma3, dbeta3, dres = _C.bn_bwd(gy2d, y, z3, mean3,
                                                   rstd3, gamma3, True,
                                                   True, True)

        # conv3 dgrad + fused bn2 backward-reduce (mask from a2, zhat
        # from z2) — gp2 is the relu-masked da2
        if w3t is None:
            w3t = w3b.t().contiguous()
        wsr = _ws_pair(z2.device, "dgrad", z2.shape[1])
        gp2, dbeta2, dgamma2 = _C.gemm_nt_dgrad_epi(dz3, w3t, a2, z2,
                                                    mean2, rstd2, None,
                                                    wsr[0], wsr[1])
        dz2 = _C.bn_bwd_dx_only(gp2, z2, mean2, rstd2, gamma2, dbeta2,
                                dgamma2)
        # conv3 wgrad over the materialized a2 (MIOpen wrw igemm: measured
        # ~6x the throughput of a hipBLASLt TN GEMM on these skinny
        # huge-K shapes)
        dw3 = _conv1x1_wgrad(dz3, a2, (N, Ho, Wo), w3_strides)

        # conv2 backward (MIOpen)
        dz2_4d = _as4d(dz2, N, Ho, Wo)
        a1_4d = _as4d(a1, N, H, W)
        da1_4d, dw2, _ = torch.ops.aten.convolution_backward(
            dz2_4d, a1_4d, w2b, None, [stride, stride], [1, 1], [1, 1],
            False, [0, 0], groups, [True, True, False])
        da1 = _as2d(da1_4d.contiguous(memory_format=torch.channels_last))

        # bn1 backward (full)
        dz1, dgamma1, dbeta1 = _C.bn_bwd(da1, a1, z1, mean1, rstd1, gamma1,
                                         True, True, False)

        # conv1 dgrad.  Identity-residual blocks fold the residual grad
        # into the store tail (c_add); when the PRECEDING block is a fused
        # bottleneck, the same epilogue also applies ITS bn3 relu mask
        # (from x = that block's output) and reduces its dbeta3/dgamma3
        # (deposited into prev_hand), so the grad handed upstream is
        # already masked and that block skips its whole reduce pass.
        if w1t is None:
            w1t = w1b.t().contiguous()
        x2d = _as2d(x)
        if has_prev and not has_sep_identity and ctx.prev_hand is not None:
            wsp = _ws_pair(x.device, "dgrad", x2d.shape[1])
            dx2d, db3p, dg3p = _C.gemm_nt_dgrad_epi(
                dz1, w1t, x2d, z3p, mean3p, rstd3p, dres, wsp[0], wsp[1])
            ctx.prev_hand["deposit"] = (db3p, dg3p)
        else:
            dx2d = _C.gemm_nt_bf16(dz1, w1t,
                                   None if has_sep_identity else dres)
        dw1 = _conv1x1_wgrad(dz1, x2d, (N, H, W), w1_strides)

        dx = _as4d(dx2d, N, H, W)
        dident = _as4d(dres, N, Ho, Wo) if has_sep_identity else None

        return (dx, dident, dw1, dgamma1, dbeta1, dw2.float(), dgamma2,
                dbeta2, dw3, dgamma3, dbeta3, None, None, None, None, None,
                None, None)


def _pow2_channels_ok(c: int) -> bool:
    return c >= 64 and (c & (c - 1)) == 0 and c <= 2048


def fused_bottleneck_applicable(x: torch.Tensor, block) -> bool:
    if not (x.is_cuda and x.dtype == torch.bfloat16 and has_ext()
            and not force_fallback()):
        return False
    if not block.bn1.training:  # eval path stays on the generic bn_act route
        return False
    if not x.is_contiguous(memory_format=torch.channels_last):
        return False
    N, Ci, H, W = x.shape
    width = block.conv1.weight.shape[0]
    co3 = block.conv3.weight.shape[0]
    stride = block.conv2.stride[0]
    Ho = (H - 1) // stride + 1
    Wo = (W - 1) // stride + 1
    if (N * H * W) % 128 != 0 or (N * Ho * Wo) % 128 != 0:
        return False
    if Ci % 64 != 0:
        return False
    return _pow2_channels_ok(width) and _pow2_channels_ok(co3)


def fused_bottleneck(x: torch.Tensor, block) -> Optional[torch.Tensor]:
    """Run a Bottleneck block through the fused conv+BN path, or return
    None when the input/config doesn't qualify (caller falls back to the
    generic per-op route)."""
    if not fused_bottleneck_applicable(x, block):
        return None
    if block.downsample is not None:
        from .bn import bn_act
        identity = bn_act(block.downsample[0](x), block.downsample[1],
                          relu=False)
    else:
        identity = None
    for bn in (block.bn1, block.bn2, block.bn3):
        if bn.num_batches_tracked is not None \
                and not getattr(bn, "_nbt_managed", False):
            bn.num_batches_tracked += 1
    bn_state = (block.bn1.running_mean, block.bn1.running_var,
                block.bn2.running_mean, block.bn2.running_var,
                block.bn3.running_mean, block.bn3.running_var,
                (block.bn1.eps, block.bn2.eps, block.bn3.eps),
                block.bn1.momentum)
    conv2_meta = (block.conv2.stride[0], block.conv2.groups)
    need_grad = torch.is_grad_enabled() and (
        x.requires_grad or block.conv1.weight.requires_grad)

    # cross-block bn3-backward fusion: when the PRECEDING bottleneck ran
    # fused this step (its state sink's output IS our input) and we have
    # an identity residual, our conv1-dgrad epilogue can take over its
    # bn3 reduce + relu mask.
    prev_bn3 = prev_hand = None
    prev = getattr(block, "_prev_bottleneck", None)
    if (need_grad and identity is None and prev is not None):
        st = getattr(prev, "_bn3_fwd_state", None)
        if st is not None and st.get("y") is x:
            prev_bn3 = (st["z3"], st["mean3"], st["rstd3"])
            prev_hand = getattr(prev, "_bn3_hand", None)
            prev._bn3_fwd_state = None  # consumed; don't retain activations
    state_sink = {} if need_grad else None
    my_hand = {} if need_grad else None
    out = _FusedBottleneck.apply(
        x, identity, block.conv1.weight, block.bn1.weight, block.bn1.bias,
        block.conv2.weight, block.bn2.weight, block.bn2.bias,
        block.conv3.weight, block.bn3.weight, block.bn3.bias,
        bn_state, conv2_meta, need_grad, prev_bn3, prev_hand, state_sink,
        my_hand)
    if need_grad:
        block._bn3_fwd_state = state_sink
        block._bn3_hand = my_hand
    return out
