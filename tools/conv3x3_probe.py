"""A/B the implicit-GEMM 3x3 conv (and the strided 1x1) vs MIOpen/CK per
ResNet-50 shape — the round-2 bring-up measurement.  Run on the GPU box:

    python tools/conv3x3_probe.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from moco_amd import _C

torch.backends.cudnn.benchmark = True
N = 256

C3_SHAPES = [  # (Ci, H, stride) for the 3x3 convs of R50 at bs256
    (64, 56, 1), (128, 56, 2), (128, 28, 1), (256, 28, 2), (256, 14, 1),
    (512, 14, 2), (512, 7, 1),
]
DS_SHAPES = [  # (Ci, H, Co) strided-1x1 downsample convs
    (256, 56, 512), (512, 28, 1024), (1024, 14, 2048),
]


def bench(fn, iters=15):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / iters


print(f"{'3x3 shape':22s} {'miopen_us':>10s} {'ours_us':>9s} {'speedup':>8s} {'maxerr':>8s}")
tm_tot = to_tot = 0.0
for Ci, H, s in C3_SHAPES:
    Co = Ci
    x = (torch.randn(N, Ci, H, H, device='cuda', dtype=torch.bfloat16) * 0.5) \
        .contiguous(memory_format=torch.channels_last)
    w = (torch.randn(Co, Ci, 3, 3, device='cuda', dtype=torch.bfloat16) * 0.04) \
        .contiguous(memory_format=torch.channels_last)
    (ours,) = _C.conv3x3_fwd(x, w, s, None, None, False, False, None)
    ref = F.conv2d(x, w, stride=s, padding=1)
    err = (ours.float() - ref.float()).abs().max().item()
    tm = bench(lambda: F.conv2d(x, w, stride=s, padding=1))
    to = bench(lambda: _C.conv3x3_fwd(x, w, s, None, None, False, False,
                                      None))
    tm_tot += tm
    to_tot += to
    print(f"{str((Ci,H,s)):22s} {tm*1e6:10.1f} {to*1e6:9.1f} {tm/to:8.2f} "
          f"{err:8.4f}")
print(f"{'TOTAL':22s} {tm_tot*1e6:10.1f} {to_tot*1e6:9.1f} "
      f"{tm_tot/to_tot:8.2f}")

print(f"\n{'1x1/s2 shape':22s} {'miopen_us':>10s} {'ours_us':>9s} {'speedup':>8s} {'maxerr':>8s}")
for Ci, H, Co in DS_SHAPES:
    x = (torch.randn(N, Ci, H, H, device='cuda', dtype=torch.bfloat16) * 0.5) \
        .contiguous(memory_format=torch.channels_last)
    w = (torch.randn(Co, Ci, 1, 1, device='cuda', dtype=torch.bfloat16) * 0.04) \
        .contiguous(memory_format=torch.channels_last)
    ours = _C.conv1x1_strided(x, w, 2)
    ref = F.conv2d(x, w, stride=2)
    err = (ours.float() - ref.float()).abs().max().item()
    tm = bench(lambda: F.conv2d(x, w, stride=2))
    to = bench(lambda: _C.conv1x1_strided(x, w, 2))
    print(f"{str((Ci,H,Co)):22s} {tm*1e6:10.1f} {to*1e6:9.1f} {tm/to:8.2f} "
          f"{err:8.4f}")
