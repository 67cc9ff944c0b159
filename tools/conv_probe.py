"""Probe MIOpen conv performance per ResNet-50 layer shape on MI355X.

Times fwd / dgrad / wgrad separately for every distinct conv shape in
ResNet-50 at a given batch, bf16 + channels_last, with cudnn.benchmark on.
Prints a table with achieved TFLOP/s so the slow shapes (naive fallbacks)
are obvious.  Run on the GPU box:  python tools/conv_probe.py [batch]
"""

import sys
import time

import torch
import torch.nn.functional as F

torch.backends.cudnn.benchmark = True

# (Cin, H, W, Cout, k, stride) — distinct convs of ResNet-50 at 224^2
R50_SHAPES = [
    (3, 224, 224, 64, 7, 2),
    (64, 56, 56, 64, 1, 1),
    (64, 56, 56, 64, 3, 1),
    (64, 56, 56, 256, 1, 1),
    (256, 56, 56, 64, 1, 1),
    (256, 56, 56, 128, 1, 1),
    (128, 56, 56, 128, 3, 2),
    (128, 28, 28, 512, 1, 1),
    (256, 56, 56, 512, 1, 2),
    (512, 28, 28, 128, 1, 1),
    (128, 28, 28, 128, 3, 1),
    (512, 28, 28, 256, 1, 1),
    (256, 28, 28, 256, 3, 2),
    (256, 14, 14, 1024, 1, 1),
    (512, 28, 28, 1024, 1, 2),
    (1024, 14, 14, 256, 1, 1),
    (256, 14, 14, 256, 3, 1),
    (1024, 14, 14, 512, 1, 1),
    (512, 14, 14, 512, 3, 2),
    (512, 7, 7, 2048, 1, 1),
    (1024, 14, 14, 2048, 1, 2),
    (2048, 7, 7, 512, 1, 1),
    (512, 7, 7, 512, 3, 1),
]


def bench(fn, iters=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / iters


def main():
    N = int(sys.argv[1]) if len(sys.argv) > 1 else 256
    dev = "cuda"
    print(f"batch={N} bf16 channels_last  (times in ms, rates in TF/s)")
    print(f"{'shape':38s} {'fwd':>8s}{'tf':>6s} {'dgrad':>8s}{'tf':>6s} "
          f"{'wgrad':>8s}{'tf':>6s}")
    tot_f = tot_d = tot_w = 0.0
    for (C, H, W, K, k, s) in R50_SHAPES:
        x = torch.randn(N, C, H, W, device=dev, dtype=torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        w = torch.randn(K, C, k, k, device=dev, dtype=torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        pad = k // 2 if k > 1 else 0
        y = F.conv2d(x, w, stride=s, padding=pad)
        g = torch.randn_like(y)
        Ho, Wo = y.shape[2], y.shape[3]
        flops = 2.0 * N * K * C * k * k * Ho * Wo

        tf = bench(lambda: F.conv2d(x, w, stride=s, padding=pad))
        xr = x.detach().requires_grad_(True)
        yy = F.conv2d(xr, w, stride=s, padding=pad)
        td = bench(lambda: torch.autograd.grad(yy, xr, g, retain_graph=True))
        wr = w.detach().requires_grad_(True)
        yw = F.conv2d(x, wr, stride=s, padding=pad)
        tw = bench(lambda: torch.autograd.grad(yw, wr, g, retain_graph=True))

        tot_f += tf
        tot_d += td
        tot_w += tw
        print(f"{str((C,H,W,K,k,s)):38s} {tf*1e3:8.3f}{flops/tf/1e12:6.0f} "
              f"{td*1e3:8.3f}{flops/td/1e12:6.0f} "
              f"{tw*1e3:8.3f}{flops/tw/1e12:6.0f}")
    print(f"{'TOTAL (1x each)':38s} {tot_f*1e3:8.3f}       "
          f"{tot_d*1e3:8.3f}       {tot_w*1e3:8.3f}")


if __name__ == "__main__":
    main()
