"""A/B the custom MFMA 1x1-conv GEMM vs MIOpen/CK per ResNet-50 shape.
Run on the GPU box: python tools/conv1x1_probe.py"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from moco_amd import _C

torch.backends.cudnn.benchmark = True

SHAPES = [  # (Ci, H, Co) for 1x1 s1 convs of R50 at bs256
    (64, 56, 256), (256, 56, 64), (256, 56, 128),
    (512, 28, 128), (128, 28, 512), (512, 28, 256),
    (1024, 14, 256), (256, 14, 1024), (1024, 14, 512),
    (2048, 7, 512), (512, 7, 2048),
]
N = 256


def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / iters


print(f"{'shape':22s} {'miopen_us':>10s} {'ours_us':>9s} {'speedup':>8s} {'maxerr':>8s}")
tot_m = tot_o = 0.0
for Ci, H, Co in SHAPES:
    M = N * H * H
    x = torch.randn(N, Ci, H, H, device='cuda', dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    w = torch.randn(Co, Ci, 1, 1, device='cuda', dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last) * 0.05
    x2d = x.permute(0, 2, 3, 1).reshape(M, Ci)
    w2d = w.view(Co, Ci).contiguous()

    ref = F.conv2d(x, w)
    ours2d = _C.gemm_nt_bf16(x2d.contiguous(), w2d)
    ours = ours2d.view(N, H, H, Co).permute(0, 3, 1, 2)
    err = (ours.float() - ref.float()).abs().max().item()

    tm = bench(lambda: F.conv2d(x, w))
    to = bench(lambda: _C.gemm_nt_bf16(x2d, w2d))
    tot_m += tm
    tot_o += to
    print(f"{str((Ci,H,Co)):22s} {tm*1e6:10.1f} {to*1e6:9.1f} "
          f"{tm/to:8.2f} {err:8.4f}")
print(f"{'TOTAL':22s} {tot_m*1e6:10.1f} {tot_o*1e6:9.1f} {tot_m/tot_o:8.2f}")
