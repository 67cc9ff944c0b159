"""On-GPU two-crop augmentation pipeline.

Replaces the reference's CPU/PIL pipeline (moco/loader.py TwoCropsTransform +
GaussianBlur, main_moco.py:237-262 aug lists) with batched HIP kernels:
crop-resize (bilinear) -> color jitter -> grayscale -> separable Gaussian
blur -> horizontal flip -> normalize, producing both crops of every image in
one pass over the batch.

Randomness: per-(image, crop) parameters are sampled on the host with torch's
RNG into a small [2B, 12] float tensor; the kernels are deterministic given
that tensor, which makes the GPU path unit-testable against the pure-torch
reference implementation below (`_augment_torch`).

Fidelity notes vs the reference transforms:
- ColorJitter sub-op order is fixed (brightness, contrast, saturation, hue)
  instead of torchvision's random permutation.
- Gaussian blur = truncated Gaussian, radius ceil(2.5*sigma), on the resized
  224x224 crop (the reference blurs the cropped image too).
- v1 order (grayscale before jitter, jitter always applied, no hue) and
  v2 order (jitter p=0.8 first, grayscale, blur p=0.5) are both supported
  via the per-image `order` flag.
"""

import math
from typing import Optional, Tuple

import torch

from . import use_hip, ext

IMAGENET_MEAN = (0.485, 0.456, 0.406)
IMAGENET_STD = (0.229, 0.224, 0.225)

# param vector layout (float32, one row per (crop, image)):
P_X0, P_Y0, P_W, P_H, P_FLIP, P_GRAY, P_SIGMA, P_BR, P_CO, P_SA, P_HU, P_ORDER = range(12)
# P_ORDER: 0 = v2 order (jitter then grayscale), 1 = v1 order (grayscale first)

GRAY_R, GRAY_G, GRAY_B = 0.299, 0.587, 0.114


def sample_two_crop_params(batch: int, src_h: int, src_w: int, out_size: int,
                           aug_plus: bool,
                           generator: Optional[torch.Generator] = None,
                           device: str = "cpu",
                           scale=(0.2, 1.0)) -> torch.Tensor:
    """Sample augmentation parameters for both crops of a batch.

    Returns [2*batch, 12] float32 (crop-major: rows [0:B] = crop q,
    rows [B:2B] = crop k).  `scale` is the RandomResizedCrop area range:
    (0.2, 1.0) for MoCo pretraining (reference main_moco.py:238,253),
    (0.08, 1.0) = torchvision default for the lincls train transform.
    """
    n = 2 * batch
    g = generator

    def rand(*shape):
        return torch.rand(*shape, generator=g)

    # RandomResizedCrop(out, scale, ratio=(3/4, 4/3)) - 10 trials
    area = float(src_h * src_w)
    log_r = (math.log(3.0 / 4.0), math.log(4.0 / 3.0))
    w = torch.empty(n)
    h = torch.empty(n)
    ok = torch.zeros(n, dtype=torch.bool)
    s_lo, s_hi = scale
    for _ in range(10):
        target_area = area * (s_lo + (s_hi - s_lo) * rand(n))
        aspect = torch.exp(log_r[0] + (log_r[1] - log_r[0]) * rand(n))
        tw = torch.sqrt(target_area * aspect).round()
        th = torch.sqrt(target_area / aspect).round()
        valid = (~ok) & (tw > 0) & (th > 0) & (tw <= src_w) & (th <= src_h)
        w[valid] = tw[valid]
        h[valid] = th[valid]
        ok |= valid
        if bool(ok.all()):
            break
    # fallback: center crop of min side
    side = float(min(src_h, src_w))
    w[~ok] = side
    h[~ok] = side
    x0 = (rand(n) * (src_w - w + 1)).floor()
    y0 = (rand(n) * (src_h - h + 1)).floor()

    p = torch.zeros(n, 12)
    p[:, P_X0] = x0
    p[:, P_Y0] = y0
    p[:, P_W] = w
    p[:, P_H] = h
    p[:, P_FLIP] = (rand(n) < 0.5).float()
    p[:, P_GRAY] = (rand(n) < 0.2).float()

    def jitter_factor(mag):
        return 1.0 + (2.0 * rand(n) - 1.0) * mag

    if aug_plus:  # MoCo v2 (main_moco.py:237-249)
        apply_j = (rand(n) < 0.8).float()
        p[:, P_BR] = 1.0 + apply_j * (jitter_factor(0.4) - 1.0)
        p[:, P_CO] = 1.0 + apply_j * (jitter_factor(0.4) - 1.0)
        p[:, P_SA] = 1.0 + apply_j * (jitter_factor(0.4) - 1.0)
        p[:, P_HU] = apply_j * (2.0 * rand(n) - 1.0) * 0.1
        blur = (rand(n) < 0.5).float()
        p[:, P_SIGMA] = blur * (0.1 + 1.9 * rand(n))
        p[:, P_ORDER] = 0.0
    else:  # MoCo v1 (main_moco.py:251-262)
        p[:, P_BR] = jitter_factor(0.4)
        p[:, P_CO] = jitter_factor(0.4)
        p[:, P_SA] = jitter_factor(0.4)
        p[:, P_HU] = 0.0
        p[:, P_SIGMA] = 0.0
        p[:, P_ORDER] = 1.0
    return p.to(device)


def two_crop_augment(images: torch.Tensor, out_size: int = 224,
                     aug_plus: bool = False,
                     params: Optional[torch.Tensor] = None,
                     out_dtype: Optional[torch.dtype] = None,
                     channels_last: bool = True,
                     generator: Optional[torch.Generator] = None
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
    """images: [B, 3, H, W] float in [0,1]. Returns (im_q, im_k), each
    [B, 3, out, out], normalized with ImageNet stats."""
    B, C, H, W = images.shape
    assert C == 3
    if params is None:
        params = sample_two_crop_params(B, H, W, out_size, aug_plus,
                                        generator=generator,
                                        device=str(images.device))
    if out_dtype is None:
        out_dtype = (images.dtype if images.dtype != torch.uint8
                     else torch.float32)
    if use_hip(images):
        out = ext().two_crop_augment(images.contiguous(), params.contiguous(),
                                     out_size,
                                     out_dtype == torch.bfloat16,
                                     channels_last)
    else:
        if images.dtype == torch.uint8:
            images = images.float() / 255.0
        out = _augment_torch(images, params, out_size).to(out_dtype)
        if channels_last:
            out = out.contiguous(memory_format=torch.channels_last)
    return out[:B], out[B:]


# ---------------------------------------------------------------------------
# Pure-torch reference (CPU path + GPU test oracle)
# ---------------------------------------------------------------------------

def _bilinear_crop_resize(images: torch.Tensor, params: torch.Tensor,
                          S: int) -> torch.Tensor:
    """Batched bilinear resize of per-image crop rects. Matches the HIP
    kernel's sampling: sx = x0 + (j+0.5)*w/S - 0.5, edge-clamped."""
    B, C, H, W = images.shape
    n = params.shape[0]
    src = images.float()
    bidx = torch.arange(n) % B
    jj = torch.arange(S, dtype=torch.float32)
    sx = params[:, P_X0, None] + (jj[None] + 0.5) * params[:, P_W, None] / S - 0.5
    sy = params[:, P_Y0, None] + (jj[None] + 0.5) * params[:, P_H, None] / S - 0.5
    x0 = sx.floor()
    y0 = sy.floor()
    fx = (sx - x0)[:, None, None, :]          # [n,1,1,S]
    fy = (sy - y0)[:, None, :, None]          # [n,1,S,1]
    x0 = x0.long()
    y0 = y0.long()
    x0c = x0.clamp(0, W - 1)
    x1c = (x0 + 1).clamp(0, W - 1)
    y0c = y0.clamp(0, H - 1)
    y1c = (y0 + 1).clamp(0, H - 1)
    b = bidx[:, None, None]                   # [n,1,1]
    # gather four corners: result [n, C, S(y), S(x)]
    def at(yi, xi):
        return src[b, :, yi[:, :, None], xi[:, None, :]].permute(0, 3, 1, 2)
    v00 = at(y0c, x0c)
    v01 = at(y0c, x1c)
    v10 = at(y1c, x0c)
    v11 = at(y1c, x1c)
    top = v00 * (1 - fx) + v01 * fx
    bot = v10 * (1 - fx) + v11 * fx
    return top * (1 - fy) + bot * fy


def _rgb_to_hsv(r, g, b):
    maxc = torch.maximum(torch.maximum(r, g), b)
    minc = torch.minimum(torch.minimum(r, g), b)
    v = maxc
    cr = maxc - minc
    s = torch.where(maxc > 0, cr / maxc.clamp_min(1e-12),
                    torch.zeros_like(maxc))
    crs = cr.clamp_min(1e-12)
    rc = (maxc - r) / crs
    gc = (maxc - g) / crs
    bc = (maxc - b) / crs
    h = torch.where(maxc == r, bc - gc,
                    torch.where(maxc == g, 2.0 + rc - bc, 4.0 + gc - rc))
    h = torch.where(cr > 0, h, torch.zeros_like(h))
    h = (h / 6.0) % 1.0
    return h, s, v


def _hsv_to_rgb(h, s, v):
    i = (h * 6.0).floor()
    f = h * 6.0 - i
    p = v * (1.0 - s)
    q = v * (1.0 - s * f)
    t = v * (1.0 - s * (1.0 - f))
    i = i.long() % 6
    r = torch.where(i == 0, v, torch.where(i == 1, q, torch.where(
        i == 2, p, torch.where(i == 3, p, torch.where(i == 4, t, v)))))
    g = torch.where(i == 0, t, torch.where(i == 1, v, torch.where(
        i == 2, v, torch.where(i == 3, q, torch.where(i == 4, p, p)))))
    b = torch.where(i == 0, p, torch.where(i == 1, p, torch.where(
        i == 2, t, torch.where(i == 3, v, torch.where(i == 4, v, q)))))
    return r, g, b


def _apply_jitter(x: torch.Tensor, params: torch.Tensor) -> torch.Tensor:
    """brightness, contrast, saturation, hue — fixed order, per image."""
    br = params[:, P_BR, None, None, None]
    co = params[:, P_CO, None, None, None]
    sa = params[:, P_SA, None, None, None]
    hu = params[:, P_HU]
    # contrast blends toward the crop's gray mean; like the HIP kernel we use
    # the PRE-brightness mean scaled by the brightness factor (exact for
    # unclipped pixels) so the single gray-mean reduction of pass 1 suffices
    gray_pre = (GRAY_R * x[:, 0] + GRAY_G * x[:, 1] + GRAY_B * x[:, 2])
    mean_g = (gray_pre.mean(dim=(1, 2))[:, None, None, None]
              * br).clamp(0, 1)
    x = (x * br).clamp(0, 1)
    x = (co * x + (1 - co) * mean_g).clamp(0, 1)
    gray = (GRAY_R * x[:, 0] + GRAY_G * x[:, 1] + GRAY_B * x[:, 2])[:, None]
    x = (sa * x + (1 - sa) * gray).clamp(0, 1)
    if bool((hu != 0).any()):
        h, s, v = _rgb_to_hsv(x[:, 0], x[:, 1], x[:, 2])
        h = (h + hu[:, None, None]) % 1.0
        r, g, b = _hsv_to_rgb(h, s, v)
        x = torch.stack([r, g, b], dim=1)
    return x


def _apply_grayscale(x: torch.Tensor, params: torch.Tensor) -> torch.Tensor:
    gray_flag = params[:, P_GRAY, None, None, None]
    gray = (GRAY_R * x[:, 0] + GRAY_G * x[:, 1] + GRAY_B * x[:, 2])[:, None]
    return torch.where(gray_flag > 0, gray.expand_as(x), x)


def _apply_blur(x: torch.Tensor, params: torch.Tensor) -> torch.Tensor:
    sigmas = params[:, P_SIGMA]
    out = x.clone()
    for i in range(x.shape[0]):
        sig = float(sigmas[i])
        if sig <= 0:
            continue
        R = max(1, math.ceil(2.5 * sig))
        t = torch.arange(-R, R + 1, dtype=torch.float32, device=x.device)
        k = torch.exp(-0.5 * (t / sig) ** 2)
        k = k / k.sum()
        img = x[i:i + 1]
        # clamp-to-edge padding (matches the HIP kernel's edge handling)
        img = torch.nn.functional.pad(img, (R, R, 0, 0), mode="replicate")
        img = torch.nn.functional.conv2d(
            img, k.view(1, 1, 1, -1).expand(3, 1, 1, -1), groups=3)
        img = torch.nn.functional.pad(img, (0, 0, R, R), mode="replicate")
        img = torch.nn.functional.conv2d(
            img, k.view(1, 1, -1, 1).expand(3, 1, -1, 1), groups=3)
        out[i] = img[0]
    return out


def _augment_torch(images: torch.Tensor, params: torch.Tensor,
                   S: int) -> torch.Tensor:
    x = _bilinear_crop_resize(images, params, S)
    v1_first = params[:, P_ORDER] > 0
    if bool(v1_first.any()):
        xg = _apply_grayscale(x, params)
        x = torch.where(v1_first[:, None, None, None], xg, x)
    x = _apply_jitter(x, params)
    if bool((~v1_first).any()):
        xg = _apply_grayscale(x, params)
        x = torch.where(v1_first[:, None, None, None], x, xg)
    x = _apply_blur(x, params)
    flip = params[:, P_FLIP, None, None, None] > 0
    x = torch.where(flip, x.flip(-1), x)
    mean = torch.tensor(IMAGENET_MEAN, device=x.device).view(1, 3, 1, 1)
    std = torch.tensor(IMAGENET_STD, device=x.device).view(1, 3, 1, 1)
    return (x - mean) / std


def single_crop_augment(images: torch.Tensor, out_size: int,
                        params: torch.Tensor,
                        out_dtype: Optional[torch.dtype] = None,
                        channels_last: bool = False) -> torch.Tensor:
    """One augmented crop per image (linear-eval training path).  params is
    [B, 12] in the layout above; image index = row index."""
    B = images.shape[0]
    assert params.shape[0] == B
    if out_dtype is None:
        out_dtype = (images.dtype if images.dtype != torch.uint8
                     else torch.float32)
    if use_hip(images):
        out = ext().two_crop_augment(images.contiguous(),
                                     params.contiguous(), out_size,
                                     out_dtype == torch.bfloat16,
                                     channels_last)
    else:
        if images.dtype == torch.uint8:
            images = images.float() / 255.0
        out = _augment_torch(images, params, out_size).to(out_dtype)
        if channels_last:
            out = out.contiguous(memory_format=torch.channels_last)
    return out
