#!/usr/bin/env python
"""fp8 vs bf16 GEMM A/B on the ResNet-50 1x1 shapes (bs 256).

Run on the GPU box: python tools/fp8_probe.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from moco_amd import _C

N_IMG = 256
SHAPES = [  # (M, N, K): conv1/conv3 1x1 GEMMs of R50 at bs256
    (N_IMG * 56 * 56, 64, 64), (N_IMG * 56 * 56, 64, 256),
    (N_IMG * 56 * 56, 256, 64), (N_IMG * 28 * 28, 128, 512),
    (N_IMG * 28 * 28, 512, 128), (N_IMG * 14 * 14, 256, 1024),
    (N_IMG * 14 * 14, 1024, 256), (N_IMG * 7 * 7, 512, 2048),
    (N_IMG * 7 * 7, 2048, 512),
]


def bench(fn, iters=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / iters


print(f"{'M,N,K':24s} {'bf16_us':>8s} {'fp8_us':>8s} {'bf16_TF':>8s} "
      f"{'fp8_TF':>8s} {'speedup':>8s}")
tb = tf = 0.0
for M, N, K in SHAPES:
    if K % 128 != 0:
        print(f"{str((M, N, K)):24s}  (K % 128 != 0: stays bf16)")
        continue
    a = torch.randn(M, K, device="cuda").to(torch.bfloat16)
    b = torch.randn(N, K, device="cuda").to(torch.bfloat16)
    a8 = a.to(torch.float8_e4m3fn)
    b8 = b.to(torch.float8_e4m3fn)
    flops = 2.0 * M * N * K
    t_b = bench(lambda: _C.gemm_nt_bf16(a, b))
    t_f = bench(lambda: _C.gemm_nt_fp8(a8, b8, 1.0))
    tb += t_b
    tf += t_f
    print(f"{str((M, N, K)):24s} {t_b*1e6:8.1f} {t_f*1e6:8.1f} "
          f"{flops/t_b/1e12:8.0f} {flops/t_f/1e12:8.0f} {t_b/t_f:8.2f}")
print(f"{'TOTAL':24s} {tb*1e6:8.1f} {tf*1e6:8.1f} {'':8s} {'':8s} "
      f"{tb/tf:8.2f}")
