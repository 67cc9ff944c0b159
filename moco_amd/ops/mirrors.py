"""bf16 weight mirrors for the fused-block GEMMs.

The fused Bottleneck path consumes bf16 copies of the fp32 master conv
weights (plus [Ci, Co] transposed copies for the dgrad GEMMs).  Casting
per use cost ~115 launches and ~1.6 ms/step; here ONE multi-tensor cast
kernel refreshes every mirror and ONE tiled-transpose kernel refreshes
every transposed mirror, per encoder per step.

Refresh points (MoCo.forward): the query set at the top of the step
(masters were updated by the eager SGD after the previous replay), the
key set right after the EMA update.  Both are inside the captured region,
reading the live master values — correct under hipGraph replay.

Mirrors hang off the weight Parameters as plain attributes
(``_mirror_bf16`` / ``_mirror_bf16_t``); ops/block.py prefers them and
falls back to inline casts when absent (standalone-block tests).
"""

from typing import List, Optional

import torch

from . import ext, use_hip


class MirrorSet:
    """All bf16 mirrors of one encoder's Bottleneck conv weights."""

    def __init__(self, encoder: torch.nn.Module):
        from ..models.resnet import Bottleneck

        self.weights: List[torch.Tensor] = []
        self.transposed: List[torch.Tensor] = []
        for m in encoder.modules():
            if not isinstance(m, Bottleneck):
                continue
            for name in ("conv1", "conv2", "conv3"):
                w = getattr(m, name).weight
                co, ci = w.shape[0], w.shape[1]
                if name == "conv2":
                    mb = torch.empty_like(w.data, dtype=torch.bfloat16)
                else:
                    if co % 64 != 0 or ci % 64 != 0:
                        continue  # fused path won't run; skip
                    mb = torch.empty(co, ci, dtype=torch.bfloat16,
                                     device=w.device)
                    wt = torch.empty(ci, co, dtype=torch.bfloat16,
                                     device=w.device)
                    w._mirror_bf16_t = wt
                    self.transposed.append((mb, wt, co, ci))
                w._mirror_bf16 = mb
                self.weights.append((w, mb))

        device = encoder.conv1.weight.device
        # cast chunk table (64Ki chunks like the EMA/SGD kernels)
        CH = 1 << 16
        sp, dp, sz = [], [], []
        for w, mb in self.weights:
            n = w.numel()
            base_s, base_d = w.data_ptr(), mb.data_ptr()
            off = 0
            while off < n:
                c = min(CH, n - off)
                sp.append(base_s + off * 4)
                dp.append(base_d + off * 2)
                sz.append(c)
                off += c
        self._cast_sp = torch.tensor(sp, dtype=torch.int64, device=device)
        self._cast_dp = torch.tensor(dp, dtype=torch.int64, device=device)
        self._cast_sz = torch.tensor(sz, dtype=torch.int32, device=device)
        # transpose tile table (64x64 tiles)
        tsp, tdp, meta = [], [], []
        for mb, wt, co, ci in self.transposed:
            for co0 in range(0, co, 64):
                for ci0 in range(0, ci, 64):
                    tsp.append(mb.data_ptr())
                    tdp.append(wt.data_ptr())
                    meta.extend([co0, ci0, co, ci])
        self._t_sp = torch.tensor(tsp, dtype=torch.int64, device=device)
        self._t_dp = torch.tensor(tdp, dtype=torch.int64, device=device)
        self._t_meta = torch.tensor(meta, dtype=torch.int32, device=device)
        self._key = tuple(w.data_ptr() for w, _ in self.weights)

    def valid(self) -> bool:
        return self._key == tuple(w.data_ptr() for w, _ in self.weights)

    @torch.no_grad()
    def refresh(self):
        if len(self._cast_sp):
            ext().multi_cast_bf16(self._cast_sp, self._cast_dp,
                                  self._cast_sz)
        if len(self._t_sp):
            ext().multi_transpose_bf16(self._t_sp, self._t_dp, self._t_meta)


def refresh_mirrors(encoder: torch.nn.Module) -> None:
    """Build (lazily) and refresh the encoder's mirror set.  No-op on CPU
    or when the HIP extension is unavailable."""
    w = getattr(encoder, "conv1", None)
    if w is None or not use_hip(w.weight):
        return
    ms: Optional[MirrorSet] = getattr(encoder, "_mirror_set", None)
    if ms is None or not ms.valid():
        ms = MirrorSet(encoder)
        object.__setattr__(encoder, "_mirror_set", ms)
    ms.refresh()
