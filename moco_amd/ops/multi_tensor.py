"""Fused multi-tensor kernels: EMA key-encoder update and SGD-momentum step.

The reference does the EMA as a Python loop over ~161 parameter tensors
(builder.py:51-56) and the optimizer step as torch SGD (main_moco.py:205-208)
— hundreds of tiny kernel launches per step.  Here both are single fused
HIP launches over a chunk table (one workgroup per 64Ki-element chunk),
with plain-PyTorch fallbacks on CPU.
"""

from typing import Iterable, List, Sequence, Tuple

import torch

from . import use_hip, ext

CHUNK = 1 << 16


def _build_chunks(tensors_lists: Sequence[Sequence[torch.Tensor]]):
    """Build per-chunk address tables for N parallel tensor lists.

    Returns (ptr_tensors, sizes) where ptr_tensors[i] is an int64 device
    tensor of chunk base addresses for list i and sizes is an int32 device
    tensor of chunk lengths.
    """
    n_lists = len(tensors_lists)
    ptrs: List[List[int]] = [[] for _ in range(n_lists)]
    sizes: List[int] = []

    def _dense(t):
        # channels_last weights are dense too; elementwise kernels only need
        # a dense buffer with IDENTICAL strides across the parallel lists
        return (t.is_contiguous()
                or t.is_contiguous(memory_format=torch.channels_last)
                or t.is_contiguous(memory_format=torch.channels_last_3d))

    for tensors in zip(*tensors_lists):
        numel = tensors[0].numel()
        for t in tensors:
            assert t.numel() == numel and _dense(t), \
                (t.shape, t.stride(), "non-dense tensor in multi-tensor op")
            assert t.stride() == tensors[0].stride(), \
                (t.stride(), tensors[0].stride(), "stride mismatch")
        esizes = [t.element_size() for t in tensors]
        base = [t.data_ptr() for t in tensors]
        off = 0
        while off < numel:
            n = min(CHUNK, numel - off)
            for i in range(n_lists):
                ptrs[i].append(base[i] + off * esizes[i])
            sizes.append(n)
            off += n
    device = tensors_lists[0][0].device
    ptr_tensors = [torch.tensor(p, dtype=torch.int64, device=device)
                   for p in ptrs]
    size_tensor = torch.tensor(sizes, dtype=torch.int32, device=device)
    return ptr_tensors, size_tensor


class MultiTensorEMA:
    """dst_i = m * dst_i + (1 - m) * src_i for parallel parameter lists.

    Chunk tables are cached; rebuilt automatically if any data_ptr moves.
    """

    def __init__(self, dst: Iterable[torch.Tensor], src: Iterable[torch.Tensor]):
        self.dst = list(dst)
        self.src = list(src)
        assert len(self.dst) == len(self.src)
        self._cache = None
        self._cache_key = None

    def _tables(self):
        key = tuple(t.data_ptr() for t in self.dst + self.src)
        if self._cache is None or key != self._cache_key:
            (pd, ps), sz = _build_chunks([self.dst, self.src])
            self._cache = (pd, ps, sz)
            self._cache_key = key
        return self._cache

    @torch.no_grad()
    def step(self, m: float) -> None:
        if not self.dst:
            return
        if use_hip(self.dst[0]):
            pd, ps, sz = self._tables()
            ext().multi_tensor_ema(pd, ps, sz, float(m))
        else:
            torch._foreach_mul_(self.dst, m)
            torch._foreach_add_(self.dst, self.src, alpha=1.0 - m)


class FusedSGD(torch.optim.Optimizer):
    """SGD + momentum + weight decay as one fused HIP launch.

    Matches torch.optim.SGD(momentum=0.9, dampening=0, nesterov=False)
    semantics bit-for-bit on the fallback path:
        g   = grad + wd * p
        buf = mom * buf + g
        p  -= lr * buf
    (reference main_moco.py:205-208 uses exactly this optimizer).
    """

    def __init__(self, params, lr: float, momentum: float = 0.9,
                 weight_decay: float = 0.0):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._tables = {}  # group idx -> (key, tables)

    def _group_tables(self, gi: int, params, grads, bufs):
        key = tuple(t.data_ptr() for t in params + grads + bufs)
        cached = self._tables.get(gi)
        if cached is not None and cached[0] == key:
            return cached[1]
        (pp, pg, pb), sz = _build_chunks([params, grads, bufs])
        self._tables[gi] = (key, (pp, pg, pb, sz))
        return pp, pg, pb, sz

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for gi, group in enumerate(self.param_groups):
            params, grads, bufs = [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if "momentum_buffer" not in state:
                    state["momentum_buffer"] = torch.zeros_like(
                        p, memory_format=torch.preserve_format)
                g = p.grad
                if g.stride() != p.stride():
                    if p.dim() == 4 and p.is_contiguous(
                            memory_format=torch.channels_last):
                        g = g.contiguous(memory_format=torch.channels_last)
                    else:
                        g = g.contiguous()
                params.append(p)
                grads.append(g)
                bufs.append(state["momentum_buffer"])
            if not params:
                continue
            lr = group["lr"]
            mom = group["momentum"]
            wd = group["weight_decay"]
            if use_hip(params[0]):
                pp, pg, pb, sz = self._group_tables(gi, params, grads, bufs)
                ext().multi_tensor_sgd(pp, pg, pb, sz, float(lr), float(mom),
                                       float(wd))
            else:
                for p, g, b in zip(params, grads, bufs):
                    gf = g.float()
                    if wd != 0.0:
                        gf = gf.add(p.float(), alpha=wd)
                    b.mul_(mom).add_(gf)
                    p.add_(b, alpha=-lr)
        return loss
