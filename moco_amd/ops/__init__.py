"""Op dispatch for the MI355X-native MoCo engine.

On CUDA (= ROCm/HIP here) devices every hot op runs through the in-tree
``moco_amd._C`` extension (hand-written gfx950 HIP kernels).  On CPU the same
ops fall back to plain PyTorch reference implementations — those are also the
numerics oracle the GPU tests compare against.

Policy (deliberate): if a tensor is on a CUDA device and the extension is not
importable, we raise instead of silently falling back — a silent eager
fallback on the GPU box would invalidate every performance number.
Set MOCO_AMD_FORCE_FALLBACK=1 to override for debugging only.
"""

import os

import torch

_EXT = None
_EXT_ERR = None


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    try:
        from moco_amd import _C  # built in-tree by setup.py build_ext --inplace

        _EXT = _C
    except ImportError as e:  # pragma: no cover - exercised only when unbuilt
        _EXT_ERR = e
        _EXT = None
    return _EXT


_try_load()


def has_ext() -> bool:
    return _try_load() is not None


def ext():
    return _try_load()


def force_fallback() -> bool:
    return os.environ.get("MOCO_AMD_FORCE_FALLBACK", "0") == "1"


def use_hip(*tensors) -> bool:
    """True when these tensors should run on the HIP kernels."""
    on_gpu = any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if not on_gpu:
        return False
    if force_fallback():
        return False
    if not has_ext():
        raise RuntimeError(
            "moco_amd._C HIP extension is not built but a GPU tensor reached "
            "the ops layer. Build it with `python setup.py build_ext --inplace` "
            f"(import error was: {_EXT_ERR!r})")
    return True


from .functional import (  # noqa: E402
    contrastive_logits,
    ema_update_,
    l2_normalize,
    queue_enqueue_,
    softmax_ce_label0,
)
from .multi_tensor import FusedSGD, MultiTensorEMA  # noqa: E402
from .augment import two_crop_augment  # noqa: E402

__all__ = [
    "has_ext",
    "ext",
    "use_hip",
    "l2_normalize",
    "contrastive_logits",
    "softmax_ce_label0",
    "queue_enqueue_",
    "ema_update_",
    "MultiTensorEMA",
    "FusedSGD",
    "two_crop_augment",
]
