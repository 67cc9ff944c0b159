#!/usr/bin/env python
"""Summarize a rocprofv3 kernel_stats.csv into step buckets.

Usage: python tools/kernel_buckets.py <kernel_stats.csv> [steps]
Prints per-bucket total ms (divided by `steps` when given) plus the top-25
kernels by total time — small enough to ship back through gpurun_out.
"""
import csv
import sys

BUCKETS = [
    ("our gemm_nt fwd (fused conv1x1)", ["gemm_nt_lds_kernel"]),
    ("our gemm_nt naive", ["gemm_nt_bf16_kernel"]),
    ("our dgrad_epi", ["gemm_nt_dgrad_epi"]),
    ("our bn stats/apply/bwd", ["bn_stats", "bn_apply", "bn_bwd",
                                "bn_finalize"]),
    ("our pool", ["maxpool"]),
    ("our augment", ["augment", "crop_resize", "jitter", "blur"]),
    ("our moco ops", ["contrastive", "softmax_ce", "l2norm", "queue",
                      "multi_tensor", "ema"]),
    ("hipblaslt (wgrad/mlp)", ["Cijk", "custom_kernel", "TensileLibrary"]),
    ("miopen naive_conv", ["naive_conv"]),
    ("miopen conv fwd", ["igemm_fwd", "ConvFwd", "Fwd_"]),
    ("miopen conv bwd-data", ["igemm_bwd", "ConvBwd", "BwdDat"]),
    ("miopen conv wrw", ["wrw", "WrW"]),
    ("miopen batchnorm", ["Batchnorm", "BatchNorm", "bnorm"]),
    ("miopen other", ["miopen", "Miopen", "MIOpen"]),
    ("torch elementwise", ["elementwise", "vectorized", "reduce_kernel",
                           "CatArray", "index_elementwise", "fill"]),
    ("conv (CK)", ["DeviceConv", "conv_fwd", "GridwiseConv", "ck::"]),
]


def bucket_of(name):
    for label, pats in BUCKETS:
        for p in pats:
            if p in name:
                return label
    return "other"


def main():
    path = sys.argv[1]
    steps = float(sys.argv[2]) if len(sys.argv) > 2 else None
    rows = []
    with open(path) as f:
        for r in csv.DictReader(f):
            rows.append((r["Name"], int(r["Calls"]),
                         float(r["TotalDurationNs"]) / 1e6))
    total = sum(ms for _, _, ms in rows)
    per = {}
    calls = {}
    for name, c, ms in rows:
        b = bucket_of(name)
        per[b] = per.get(b, 0.0) + ms
        calls[b] = calls.get(b, 0) + c
    div = steps if steps else 1.0
    unit = "ms/step" if steps else "ms total"
    print(f"kernel time: {total / div:.2f} {unit}  ({len(rows)} kernels)")
    for b, ms in sorted(per.items(), key=lambda kv: -kv[1]):
        print(f"{ms / div:9.3f} {unit}  {calls[b] / div:7.1f} calls  {b}")
    print("\ntop kernels:")
    for name, c, ms in sorted(rows, key=lambda r: -r[2])[:25]:
        print(f"{ms / div:9.3f} {unit}  {c / div:7.1f} calls  {name[:100]}")


def window_trace(path, nsteps=2):
    """Steady-state bucket summary from a rocprofv3 kernel_trace.csv:
    windows the timeline between contrastive_fwd dispatches (1/step),
    taking the LAST `nsteps` full steps."""
    rows = []
    with open(path) as f:
        rd = csv.DictReader(f)
        kn = "Kernel_Name" if "Kernel_Name" in rd.fieldnames else "Name"
        t0n = "Start_Timestamp" if "Start_Timestamp" in rd.fieldnames \
            else "BeginNs"
        t1n = "End_Timestamp" if "End_Timestamp" in rd.fieldnames else "EndNs"
        for r in rd:
            rows.append((r[kn], int(r[t0n]), int(r[t1n])))
    marks = sorted(t0 for n, t0, _ in rows if "contrastive_fwd" in n)
    if len(marks) < nsteps + 1:
        print(f"only {len(marks)} step markers; need {nsteps + 1}")
        return
    lo, hi = marks[-1 - nsteps], marks[-1]
    span_ms = (hi - lo) / 1e6
    per, calls = {}, {}
    tot = 0.0
    for n, t0, t1 in rows:
        if lo <= t0 < hi:
            b = bucket_of(n)
            ms = (t1 - t0) / 1e6
            per[b] = per.get(b, 0.0) + ms
            calls[b] = calls.get(b, 0) + 1
            tot += ms
    print(f"window: {nsteps} steps, {span_ms / nsteps:.2f} ms/step wall, "
          f"{tot / nsteps:.2f} ms/step kernel-busy, "
          f"{(span_ms - tot) / nsteps:.2f} ms/step gaps")
    for b, ms in sorted(per.items(), key=lambda kv: -kv[1]):
        print(f"{ms / nsteps:9.3f} ms/step  {calls[b] / nsteps:7.1f} calls  "
              f"{b}")
    # top kernels inside the window
    agg = {}
    for n, t0, t1 in rows:
        if lo <= t0 < hi:
            a = agg.setdefault(n, [0, 0.0])
            a[0] += 1
            a[1] += (t1 - t0) / 1e6
    print("\ntop kernels (window):")
    for n, (c, ms) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:30]:
        print(f"{ms / nsteps:9.3f} ms/step  {c / nsteps:7.1f} calls  "
              f"{n[:110]}")


if __name__ == "__main__":
    if "--trace" in sys.argv:
        sys.argv.remove("--trace")
        window_trace(sys.argv[1],
                     int(sys.argv[2]) if len(sys.argv) > 2 else 2)
    else:
        main()
