"""MoCo: Momentum Contrast, MI355X-native implementation.

API and checkpoint-layout parity with the reference ``moco/builder.py``
(class MoCo: __init__:16, _momentum_update_key_encoder:51,
_dequeue_and_enqueue:59, _batch_shuffle_ddp:76, _batch_unshuffle_ddp:106,
forward:126, concat_all_gather:166), re-designed for the hardware:

- EMA of the key encoder is ONE fused multi-tensor HIP launch instead of a
  161-iteration Python loop.
- The contrastive logits [q.k+ | q @ queue]/T are one fused MFMA kernel; the
  queue's GEMM working copy is stored key-per-row ([K, C] bf16) so both the
  forward and backward GEMM read K-contiguous bf16 fragments.
- Shuffling-BN derives its permutation from a shared per-step seed (zero
  communication, replaces the rank-0 broadcast) and exchanges only the rows
  each peer needs (all-to-all over xGMI instead of a full all-gather).
- The canonical queue buffer keeps the reference layout/dtype
  ([dim, K] fp32, key name ``queue``; ``queue_ptr`` [1]) so checkpoints
  interoperate.

Invariants preserved (reference forward, builder.py:126-163): EMA fires
before the key forward, once per step; the queue is updated after the logits
(current keys are positives, the queue holds past batches only); gradients
flow only through q; the queue is L2-normalized at init and replicated
identically on every rank.
"""

from typing import Callable, Optional

import torch
import torch.nn as nn

from . import parallel
from .ops import (MultiTensorEMA, contrastive_logits, l2_normalize,
                  queue_enqueue_, softmax_ce_label0)


class MoCo(nn.Module):
    def __init__(self, base_encoder: Callable[..., nn.Module], dim: int = 128,
                 K: int = 65536, m: float = 0.999, T: float = 0.07,
                 mlp: bool = False, shuffle_seed: int = 0):
        super().__init__()
        self.K = K
        self.m = m
        self.T = T
        self.shuffle_seed = shuffle_seed

        self.encoder_q = base_encoder(num_classes=dim)
        self.encoder_k = base_encoder(num_classes=dim)

        if mlp:  # MoCo v2 projection head (reference builder.py:24-28)
            dim_mlp = self.encoder_q.fc.weight.shape[1]
            self.encoder_q.fc = nn.Sequential(
                nn.Linear(dim_mlp, dim_mlp), nn.ReLU(), self.encoder_q.fc)
            self.encoder_k.fc = nn.Sequential(
                nn.Linear(dim_mlp, dim_mlp), nn.ReLU(), self.encoder_k.fc)

        for param_q, param_k in zip(self.encoder_q.parameters(),
                                    self.encoder_k.parameters()):
            param_k.data.copy_(param_q.data)
            param_k.requires_grad = False

        # canonical queue: [dim, K] fp32, L2-normalized columns
        # (reference builder.py:36-38).  Drawn from a dedicated generator
        # seeded by shuffle_seed so the ring is replicated across ranks BY
        # CONSTRUCTION even when each rank seeds its global RNG differently
        # (DDP syncs parameters at wrap time but we run
        # broadcast_buffers=False, so nothing else would ever reconcile a
        # divergent initial queue).
        qgen = torch.Generator().manual_seed(0x5EED ^ shuffle_seed)
        self.register_buffer("queue",
                             nn.functional.normalize(
                                 torch.randn(dim, K, generator=qgen),
                                 dim=0))
        self.register_buffer("queue_ptr", torch.zeros(1, dtype=torch.long))

        self._ema: Optional[MultiTensorEMA] = None
        self._queue_kc: Optional[torch.Tensor] = None  # [K, dim] bf16 (GPU)
        self._ptr_host: Optional[int] = None  # avoids a per-step device sync
        self._step = 0
        # hipGraph-replay-safe stepping (set by the engine): enqueue is
        # deferred to flush_queue() after backward (no queue snapshot
        # needed; ring ptr advanced on device), and at world size 1 the
        # shuffling-BN exchange is skipped — a permutation is a no-op for
        # batch statistics and is exactly inverted by the unshuffle, so
        # single-rank semantics are unchanged (FP reduction order aside).
        self.graph_safe = False
        self._pending_keys: Optional[torch.Tensor] = None

    # -- key-encoder EMA ----------------------------------------------------
    @torch.no_grad()
    def _momentum_update_key_encoder(self):
        if self._ema is None:
            self._ema = MultiTensorEMA(list(self.encoder_k.parameters()),
                                       list(self.encoder_q.parameters()))
        self._ema.step(self.m)

    # -- queue --------------------------------------------------------------
    def _working_queue(self) -> torch.Tensor:
        """The [K, C] layout the GEMM kernels consume.

        On GPU this is a bf16 copy maintained incrementally by the enqueue
        kernel; on CPU it is just a transposed view of the fp32 queue.
        """
        if self.queue.is_cuda:
            if self._queue_kc is None or \
                    self._queue_kc.device != self.queue.device:
                self._queue_kc = self.queue.t().contiguous().to(torch.bfloat16)
            return self._queue_kc
        return self.queue.t()

    def _load_from_state_dict(self, *args, **kwargs):
        self._queue_kc = None  # rebuilt lazily from the loaded fp32 queue
        self._ema = None
        self._ptr_host = None
        super()._load_from_state_dict(*args, **kwargs)

    @torch.no_grad()
    def _dequeue_and_enqueue(self, keys: torch.Tensor):
        keys = parallel.concat_all_gather(keys)  # C6: all ranks enqueue all
        qkc = self._queue_kc if self.queue.is_cuda else None
        if self.graph_safe and self.queue.is_cuda:
            # device-side ring position: correct under hipGraph replay
            from .ops import ext
            ext().queue_enqueue_dev(self.queue, qkc, keys.contiguous(),
                                    self.queue_ptr)
            self._ptr_host = None  # host mirror invalid in graph mode
            return
        if self._ptr_host is None:
            self._ptr_host = int(self.queue_ptr.item())
        queue_enqueue_(self.queue, qkc, self.queue_ptr, keys,
                       ptr=self._ptr_host)
        self._ptr_host = (self._ptr_host + keys.shape[0]) % self.K

    @torch.no_grad()
    def flush_queue(self):
        """Perform the enqueue deferred by a graph-safe forward (call after
        backward; the queue must not change before the contrastive
        backward reads it)."""
        if self._pending_keys is not None:
            self._dequeue_and_enqueue(self._pending_keys)
            self._pending_keys = None

    # -- shuffling-BN (world-aware; see parallel/comm.py) -------------------
    @torch.no_grad()
    def _batch_shuffle(self, x: torch.Tensor, step: int):
        return parallel.shuffle_exchange(x, step, self.shuffle_seed)

    @torch.no_grad()
    def _batch_unshuffle(self, x: torch.Tensor, idx_unshuffle: torch.Tensor):
        return parallel.unshuffle_exchange(x, idx_unshuffle)

    # -- forward ------------------------------------------------------------
    def forward(self, im_q: torch.Tensor, im_k: torch.Tensor,
                step: Optional[int] = None):
        """Compute contrastive logits and labels.

        Returns (logits [N, 1+K], labels [N] all-zero) — parity with the
        reference forward (builder.py:126-163).
        """
        if step is None:
            step = self._step
            self._step += 1
        graph_safe = self.graph_safe and parallel.get_world_size() == 1
        # the shuffle permutation is derived from (seed, step) on every rank
        # independently — verify the counters agree periodically (a desynced
        # rank would otherwise corrupt the key pairing SILENTLY)
        if step % 64 == 0 and not graph_safe:
            parallel.assert_step_sync(step, im_q.device)

        with torch.no_grad():
            # start the key-batch exchange on the comm stream FIRST so the
            # xGMI transfer overlaps the query forward (join at key forward)
            if not graph_safe:
                shuffle_job = parallel.begin_shuffle(im_k, step,
                                                     self.shuffle_seed)
            # EMA before the key forward, once per step (reference invariant)
            self._momentum_update_key_encoder()
            # batched bf16 weight-mirror refresh (ops/mirrors.py): query
            # masters changed in the last optimizer step, key masters in
            # the EMA above
            from .ops.mirrors import refresh_mirrors
            refresh_mirrors(self.encoder_q)
            refresh_mirrors(self.encoder_k)

        q = self.encoder_q(im_q)
        q = l2_normalize(q, dim=1)

        with torch.no_grad():
            if graph_safe:
                # world-1 shuffle is exactly inverted by the unshuffle and
                # batch statistics are permutation-invariant: skip it (the
                # host-generated per-step permutation is not replay-safe)
                k = self.encoder_k(im_k)
                k = l2_normalize(k, dim=1)
            else:
                im_k_sh, idx_unshuffle = shuffle_job.wait()
                k = self.encoder_k(im_k_sh)
                k = l2_normalize(k, dim=1)
                k = self._batch_unshuffle(k, idx_unshuffle)

        if q.is_cuda and (q.dtype == torch.bfloat16
                          or torch.is_autocast_enabled()):
            # bf16 MFMA path; an explicit fp32 run keeps fp32 precision and
            # takes the hipBLASLt fallback inside contrastive_logits
            q = q.to(torch.bfloat16)
            k = k.to(torch.bfloat16)
        if graph_safe:
            # enqueue deferred past backward: the queue is stable for the
            # whole fwd+bwd — no snapshot at all
            logits = contrastive_logits(q, k, self._working_queue(),
                                        self.queue, self.T, stable=True)
            labels = torch.zeros(logits.shape[0], dtype=torch.long,
                                 device=logits.device)
            self._pending_keys = k.detach()
            return logits, labels
        # tell the op which queue block the enqueue below will overwrite so
        # its backward snapshots only those columns (not the whole ring)
        if self._ptr_host is None:
            self._ptr_host = int(self.queue_ptr.item())
        span = k.shape[0] * parallel.get_world_size()
        logits = contrastive_logits(q, k, self._working_queue(), self.queue,
                                    self.T, snap_ptr=self._ptr_host,
                                    snap_span=span)
        labels = torch.zeros(logits.shape[0], dtype=torch.long,
                             device=logits.device)
        self._dequeue_and_enqueue(k)
        return logits, labels


def moco_loss(logits: torch.Tensor) -> torch.Tensor:
    """InfoNCE with the positive at index 0 — fused softmax-CE."""
    return softmax_ce_label0(logits)
