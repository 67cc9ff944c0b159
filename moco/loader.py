"""Reference-compatible import path (`from moco import loader`;
upstream moco/loader.py).  The implementation lives in moco_amd.loader."""

from moco_amd.loader import GaussianBlur, TwoCropsTransform  # noqa: F401

__all__ = ["TwoCropsTransform", "GaussianBlur"]
