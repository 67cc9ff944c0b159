"""Autograd-wrapped hot ops.

Each op has a HIP path (moco_amd._C, gfx950 MFMA/LDS kernels) and a pure
PyTorch fallback used on CPU and as the test oracle.

Reference parity notes (upstream facebookresearch/moco layout):
- l2_normalize      <- nn.functional.normalize at builder.py:131,138
- contrastive_logits<- einsum('nc,nc->n') + einsum('nc,ck->nk') + /T,
                       builder.py:146-155, fused into one kernel pair here
- softmax_ce_label0 <- nn.CrossEntropyLoss with labels==0, main_moco.py:202,310
- queue_enqueue_    <- builder.py:59-74 (_dequeue_and_enqueue)
"""

from typing import Optional, Tuple

import torch

from . import use_hip, ext


# ---------------------------------------------------------------------------
# L2 normalize (row-wise, dim=1)
# ---------------------------------------------------------------------------

def _l2norm_shape_ok(x: torch.Tensor) -> bool:
    return x.shape[1] <= 512  # one wave per row, <= 8 elements per lane


class _L2NormalizeRows(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor) -> torch.Tensor:
        if use_hip(x) and _l2norm_shape_ok(x):
            y, inv_norm = ext().l2norm_rows_fwd(x.contiguous())
        else:
            x = x.contiguous()
            xf = x.float()
            norm = xf.norm(dim=1, keepdim=True).clamp_min(1e-12)
            inv_norm = (1.0 / norm).squeeze(1)
            y = (xf * inv_norm.unsqueeze(1)).to(x.dtype)
        ctx.save_for_backward(y, inv_norm)
        return y

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        y, inv_norm = ctx.saved_tensors
        if use_hip(grad_out) and _l2norm_shape_ok(grad_out):
            gx = ext().l2norm_rows_bwd(grad_out.contiguous(), y, inv_norm)
        else:
            g = grad_out.float()
            yf = y.float()
            dot = (g * yf).sum(dim=1, keepdim=True)
            gx = ((g - yf * dot) * inv_norm.unsqueeze(1)).to(grad_out.dtype)
        return gx


def l2_normalize(x: torch.Tensor, dim: int = 1) -> torch.Tensor:
    """L2-normalize along `dim`. dim=1 (rows) is the hot path."""
    if dim == 1 and x.dim() == 2:
        return _L2NormalizeRows.apply(x)
    # cold paths (e.g. queue init along dim=0): plain torch
    return torch.nn.functional.normalize(x, dim=dim)


# ---------------------------------------------------------------------------
# Fused contrastive logits: [q.k+ | q @ queue] / T
# ---------------------------------------------------------------------------

class _ContrastiveLogits(torch.autograd.Function):
    """logits[n, 0]   = (q[n] . k[n]) / T
       logits[n, 1+j] = (q[n] . queue[:, j]) / T

    Gradients flow to q only (k and queue are detached by construction:
    reference builder.py:133-139,149).

    Layouts (MFMA NT-GEMMs want the CONTRACTION index contiguous in both
    operands — see ops/hip/contrastive.hip):
    - forward contracts over C: uses queue_kc [K, C] (key-per-row, bf16 on
      GPU) so q rows and queue rows both read C-contiguous fragments;
    - backward contracts over K: uses queue_ck [C, K] (the canonical fp32
      ring buffer) so grad rows and queue_ck rows both read K-contiguous
      fragments.
    """

    @staticmethod
    def forward(ctx, q: torch.Tensor, k: torch.Tensor,
                queue_kc: torch.Tensor, queue_ck: torch.Tensor,
                T: float, snap_ptr: Optional[int] = None,
                snap_span: Optional[int] = None,
                stable: bool = False) -> torch.Tensor:
        q = q.contiguous()
        k = k.contiguous()
        # MFMA kernel constraints (non-standard dims run the hipBLASLt
        # einsum path on GPU instead — a documented shape fallback, not a
        # silent downgrade of the supported path)
        C, K = q.shape[1], queue_kc.shape[0]
        shapes_ok = (C % 32 == 0 and C <= 512 and K % 32 == 0
                     and q.dtype == torch.bfloat16)
        ctx.shapes_ok = shapes_ok
        if use_hip(q) and shapes_ok:
            logits = ext().contrastive_fwd(q, k, queue_kc.contiguous(),
                                           1.0 / T)
        else:
            # negatives from the canonical fp32 queue (queue_ck), NOT the
            # bf16 working copy: an explicit fp32 run must keep fp32
            # precision, and the backward below contracts against queue_ck
            # — forward and backward must see the same negatives
            qf, kf = q.float(), k.float()
            l_pos = (qf * kf).sum(dim=1, keepdim=True)
            l_neg = qf @ queue_ck.float()
            logits = torch.cat([l_pos, l_neg], dim=1) / T
        # snapshot: the enqueue that follows this forward overwrites
        # `snap_span` columns of the ring buffer at `snap_ptr` in place
        # (reference dodges this with queue.clone().detach() at
        # builder.py:149, a full-queue clone every step).  When the caller
        # tells us which block the enqueue will hit, save ONLY those columns
        # (C x global-batch fp32, ~131 KB at the reference config, vs 33 MB
        # for the clone at K=65536) and splice them back around the backward
        # GEMM.  Without that info (direct op-level use), full clone.
        if stable:
            # caller guarantees the queue is NOT modified before backward
            # (deferred enqueue / hipGraph mode): no snapshot at all
            ctx.save_for_backward(k)
            ctx.queue_ck_live = queue_ck
            ctx.snap = "stable"
        elif snap_ptr is not None and snap_span is not None:
            cols = queue_ck[:, snap_ptr:snap_ptr + snap_span].clone()
            ctx.save_for_backward(k, cols)
            ctx.queue_ck_live = queue_ck  # plain attr: bypasses the version
            # counter on purpose — we restore the mutated block ourselves
            ctx.snap = (snap_ptr, snap_span)
        else:
            ctx.save_for_backward(k, queue_ck.clone())
            ctx.snap = None
        ctx.inv_T = 1.0 / T
        ctx.q_dtype = q.dtype
        return logits

    @staticmethod
    def backward(ctx, grad_logits: torch.Tensor):
        if ctx.snap == "stable":
            (k,) = ctx.saved_tensors
            saved = None
            queue_ck = ctx.queue_ck_live
        else:
            k, saved = ctx.saved_tensors
        if ctx.snap is not None and ctx.snap != "stable":
            ptr, span = ctx.snap
            queue_ck = ctx.queue_ck_live
            # splice the forward-time columns back in for the GEMM, then
            # restore the live (post-enqueue) contents.  Safe: backward runs
            # on the forward's stream before the next step's enqueue.
            cur = queue_ck[:, ptr:ptr + span].clone()
            queue_ck[:, ptr:ptr + span] = saved
        elif ctx.snap is None:
            queue_ck = saved
        if use_hip(grad_logits) and ctx.shapes_ok:
            gq = ext().contrastive_bwd(grad_logits.contiguous(), k,
                                       queue_ck.contiguous(), ctx.inv_T)
        else:
            g = grad_logits.float()
            gq = (g[:, :1] * k.float() +
                  g[:, 1:] @ queue_ck.float().t()) * ctx.inv_T
        if ctx.snap is not None and ctx.snap != "stable":
            queue_ck[:, ptr:ptr + span] = cur
        return (gq.to(ctx.q_dtype), None, None, None, None, None, None,
                None)


def contrastive_logits(q: torch.Tensor, k: torch.Tensor,
                       queue_kc: torch.Tensor, queue_ck: torch.Tensor,
                       T: float, snap_ptr: Optional[int] = None,
                       snap_span: Optional[int] = None,
                       stable: bool = False) -> torch.Tensor:
    return _ContrastiveLogits.apply(q, k, queue_kc, queue_ck, T, snap_ptr,
                                    snap_span, stable)


# ---------------------------------------------------------------------------
# Softmax cross-entropy specialized for label == 0 (fp32 logits)
# ---------------------------------------------------------------------------

class _SoftmaxCELabel0(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor) -> torch.Tensor:
        logits = logits.contiguous()
        if use_hip(logits):
            loss_rows, row_max, row_lse = ext().softmax_ce0_fwd(logits)
        else:
            lf = logits.float()
            row_max = lf.max(dim=1).values
            row_lse = torch.log(torch.exp(lf - row_max[:, None]).sum(dim=1))
            loss_rows = row_max + row_lse - lf[:, 0]
        ctx.save_for_backward(logits, row_max, row_lse)
        return loss_rows.mean()

    @staticmethod
    def backward(ctx, grad_loss: torch.Tensor):
        logits, row_max, row_lse = ctx.saved_tensors
        n = logits.shape[0]
        if use_hip(logits):
            # grad_loss stays on device (a host read would sync — and is
            # illegal under hipGraph capture); the kernel multiplies it in
            dlogits = ext().softmax_ce0_bwd(logits, row_max, row_lse,
                                            1.0 / n, grad_loss.float())
        else:
            lf = logits.float()
            p = torch.exp(lf - (row_max + row_lse)[:, None])
            p[:, 0] -= 1.0
            dlogits = p * (grad_loss / n)
        return dlogits.to(logits.dtype)


def softmax_ce_label0(logits: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy over rows, target class fixed at index 0."""
    return _SoftmaxCELabel0.apply(logits)


# ---------------------------------------------------------------------------
# Queue enqueue (ring-buffer column-block write)
# ---------------------------------------------------------------------------

@torch.no_grad()
def queue_enqueue_(queue: torch.Tensor, queue_kc: Optional[torch.Tensor],
                   queue_ptr: torch.Tensor, keys: torch.Tensor,
                   ptr: Optional[int] = None) -> None:
    """Write the gathered key batch into the ring buffer at queue_ptr.

    queue:    [C, K] fp32 — canonical, checkpoint-visible layout
              (reference builder.py:36-38,68-73)
    queue_kc: [K, C] working layout for the GEMM kernels (same values),
              or None on the CPU path.
    queue_ptr:[1] int64
    keys:     [B, C] (already gathered across ranks, L2-normalized)
    ptr:      host-side mirror of queue_ptr — pass it to avoid a per-step
              device sync (the .item() read)
    """
    B = keys.shape[0]
    K = queue.shape[1]
    assert K % B == 0, f"queue size {K} must be divisible by batch {B}"
    if ptr is None:
        ptr = int(queue_ptr.item())
    if use_hip(queue):
        ext().queue_enqueue(queue, queue_kc, keys.contiguous(), ptr)
    else:
        kf = keys.float()
        queue[:, ptr:ptr + B] = kf.t()
        if queue_kc is not None:
            queue_kc[ptr:ptr + B] = kf.to(queue_kc.dtype)
    queue_ptr[0] = (ptr + B) % K


# ---------------------------------------------------------------------------
# Fused EMA (momentum) update — convenience single-pair form
# ---------------------------------------------------------------------------

@torch.no_grad()
def ema_update_(dst: torch.Tensor, src: torch.Tensor, m: float) -> None:
    """dst = m * dst + (1 - m) * src  (reference builder.py:51-56)."""
    if use_hip(dst):
        ext().ema_flat(dst.view(-1), src.view(-1), m)
    else:
        dst.mul_(m).add_(src.to(dst.dtype), alpha=1.0 - m)
