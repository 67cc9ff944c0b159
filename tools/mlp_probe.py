#!/usr/bin/env python
"""Isolate the MoCo-v2 MLP-head GEMMs (round-3 target: ~2.2 ms/step of
hipBLASLt/CK batched kernels at M=256).  Run on the GPU box."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def bench(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / iters

M = 256
head = torch.nn.Sequential(
    torch.nn.Linear(2048, 2048), torch.nn.ReLU(),
    torch.nn.Linear(2048, 128)).cuda().to(torch.bfloat16)
x = torch.randn(M, 2048, device="cuda", dtype=torch.bfloat16,
                requires_grad=True)

t_f = bench(lambda: head(x))
def fb():
    y = head(x)
    y.sum().backward()
t_fb = bench(fb)
print(f"MLP head fwd: {t_f*1e6:.1f} us   fwd+bwd: {t_fb*1e6:.1f} us "
      f"(x2 encoders/step fwd, x1 bwd)")
for (m, n, k) in [(256, 2048, 2048), (256, 128, 2048), (2048, 2048, 256)]:
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(k, n, device="cuda", dtype=torch.bfloat16)
    t = bench(lambda: a @ b)
    print(f"matmul {m}x{n}x{k}: {t*1e6:8.1f} us  "
          f"{2*m*n*k/t/1e12:6.1f} TF")
