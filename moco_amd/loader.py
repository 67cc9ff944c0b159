"""Data pipeline.

MI355X-first design: the loader's job is to deliver RAW image tensors
([3, H, W] float in [0,1]); the two-crop augmentation itself runs on the GPU
(moco_amd.ops.two_crop_augment), not in CPU worker processes — the
reference's CPU/PIL pipeline (moco/loader.py, main_moco.py:231-268) is the
classic throughput bottleneck that design removes.

Also provided for API parity with the reference moco/loader.py:
- TwoCropsTransform (loader.py:10) — tensor-based here
- GaussianBlur     (loader.py:23) — tensor-based here

Datasets:
- SyntheticImageDataset: seeded random images (no network / no ImageNet in
  this environment; BASELINE.json prescribes synthetic data + random init).
- ImageFolderDataset: minimal ImageFolder replacement (requires PIL), for
  users with a real dataset on disk.
"""

import math
import os
import random
from typing import Callable, List, Optional, Tuple

import torch
from torch.utils.data import Dataset

from .ops.augment import (IMAGENET_MEAN, IMAGENET_STD, _apply_blur,
                          _augment_torch, sample_two_crop_params)


class TwoCropsTransform:
    """Take two random crops of one image as the query and key."""

    def __init__(self, base_transform: Callable):
        self.base_transform = base_transform

    def __call__(self, x):
        q = self.base_transform(x)
        k = self.base_transform(x)
        return [q, k]


class GaussianBlur:
    """Gaussian blur with sigma ~ U[sigma_min, sigma_max] (SimCLR-style),
    operating on [3, H, W] tensors."""

    def __init__(self, sigma: Tuple[float, float] = (0.1, 2.0)):
        self.sigma = sigma

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        sigma = random.uniform(self.sigma[0], self.sigma[1])
        params = torch.zeros(1, 12)
        params[0, 6] = sigma  # P_SIGMA
        return _apply_blur(x.unsqueeze(0), params)[0]


class SyntheticImageDataset(Dataset):
    """Deterministic random 'images' shaped like a decoded dataset.

    Each item is a raw [3, size, size] float tensor in [0,1]; augmentation is
    applied downstream (on the GPU in the training engine).
    """

    def __init__(self, length: int = 1_281_167, size: int = 256,
                 num_classes: int = 1000, seed: int = 1234):
        self.length = length
        self.size = size
        self.num_classes = num_classes
        self.seed = seed

    def __len__(self):
        return self.length

    def __getitem__(self, idx: int):
        g = torch.Generator()
        g.manual_seed(self.seed * 1_000_003 + idx)
        img = torch.rand(3, self.size, self.size, generator=g)
        label = idx % self.num_classes
        return img, label


IMG_EXTENSIONS = (".jpg", ".jpeg", ".png", ".ppm", ".bmp", ".webp")


class ImageFolderDataset(Dataset):
    """Minimal ImageFolder: root/<class>/<image>. Decodes with PIL and
    resizes the short side to `load_size`, returning [3, H, W] float."""

    def __init__(self, root: str, load_size: int = 256,
                 transform: Optional[Callable] = None):
        try:
            from PIL import Image  # noqa: F401
        except ImportError as e:
            raise RuntimeError(
                "ImageFolderDataset requires PIL; use --data synthetic "
                "in environments without it") from e
        self.root = root
        self.load_size = load_size
        self.transform = transform
        classes = sorted(d for d in os.listdir(root)
                         if os.path.isdir(os.path.join(root, d)))
        self.class_to_idx = {c: i for i, c in enumerate(classes)}
        self.samples: List[Tuple[str, int]] = []
        for c in classes:
            cdir = os.path.join(root, c)
            for fname in sorted(os.listdir(cdir)):
                if fname.lower().endswith(IMG_EXTENSIONS):
                    self.samples.append((os.path.join(cdir, fname),
                                         self.class_to_idx[c]))

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx: int):
        from PIL import Image

        path, label = self.samples[idx]
        S = self.load_size
        with Image.open(path) as im:
            im = im.convert("RGB")
            w, h = im.size
            # short side -> S, then center crop to SxS: batches must have a
            # fixed shape for the batched on-GPU augmentation downstream
            scale = S / min(w, h)
            nw, nh = max(S, round(w * scale)), max(S, round(h * scale))
            im = im.resize((nw, nh))
            left, top = (nw - S) // 2, (nh - S) // 2
            im = im.crop((left, top, left + S, top + S))
            t = torch.frombuffer(bytearray(im.tobytes()), dtype=torch.uint8)
            t = t.view(S, S, 3).permute(2, 0, 1)
        img = t.float() / 255.0
        if self.transform is not None:
            img = self.transform(img)
        return img, label


def build_dataset(data: str, load_size: int = 256, synthetic_len: int = 2048,
                  seed: int = 1234, num_classes: int = 1000) -> Dataset:
    if data == "synthetic" or data == "":
        return SyntheticImageDataset(length=synthetic_len, size=load_size,
                                     seed=seed, num_classes=num_classes)
    train_dir = os.path.join(data, "train")
    root = train_dir if os.path.isdir(train_dir) else data
    return ImageFolderDataset(root, load_size=load_size)


def cpu_two_crop_eval_transform(out_size: int = 224):
    """Deterministic center-crop + normalize for eval paths (CPU tensor in,
    CPU tensor out).  Accepts a single [3,H,W] image or a whole [B,3,H,W]
    batch (one interpolate/crop/normalize for the batch — no per-image
    Python loop)."""
    mean = torch.tensor(IMAGENET_MEAN).view(3, 1, 1)
    std = torch.tensor(IMAGENET_STD).view(3, 1, 1)

    def _t(x: torch.Tensor) -> torch.Tensor:
        single = x.dim() == 3
        if single:
            x = x.unsqueeze(0)
        h, w = x.shape[2], x.shape[3]
        scale = out_size / min(h, w)
        nh, nw = max(out_size, int(round(h * scale))), max(
            out_size, int(round(w * scale)))
        x = torch.nn.functional.interpolate(
            x, size=(nh, nw), mode="bilinear", align_corners=False)
        top = (nh - out_size) // 2
        left = (nw - out_size) // 2
        x = x[:, :, top:top + out_size, left:left + out_size]
        x = (x - mean) / std
        return x.squeeze(0) if single else x

    return _t
