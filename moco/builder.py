"""Reference-compatible import path (`from moco import builder`;
upstream moco/builder.py).  The implementation lives in moco_amd.builder."""

from moco_amd.builder import MoCo, moco_loss  # noqa: F401
from moco_amd.parallel import concat_all_gather  # noqa: F401

__all__ = ["MoCo", "moco_loss", "concat_all_gather"]
