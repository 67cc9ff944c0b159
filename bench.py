#!/usr/bin/env python
"""Driver benchmark contract: ResNet-50 MoCo-v2 pretraining step throughput.

    python bench.py --gpus N --steps K --warmup W

For N>1 the driver launches this under torch.distributed.run with one rank
per GPU over RCCL.  Rank 0 prints ONE JSON line with the whole-job img/s.

Timed region = the full training step: on-GPU two-crop augmentation ->
query/key forwards (bf16, channels_last) -> fused contrastive logits ->
fused softmax-CE -> backward (DDP all-reduce overlap) -> fused SGD step ->
queue enqueue.  Weak scaling: per-GPU batch is fixed as N grows.
"""

import argparse
import json
import os
import time

import torch
import torch.distributed as dist

import moco_amd
from moco_amd.engine import MoCoStepper
from moco_amd.models.resnet import build_model
from moco_amd.ops import FusedSGD

BASELINE_NODE_IMG_S = 1340.0  # 8x V100 derived


def get_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch-per-gpu", type=int, default=256)
    p.add_argument("--moco-k", type=int, default=65536)
    p.add_argument("--arch", default="resnet50")
    p.add_argument("--crop-size", type=int, default=224)
    p.add_argument("--src-size", type=int, default=256)
    p.add_argument("--pool-batches", type=int, default=4)
    p.add_argument("--phase-timing", action="store_true",
                   help="print per-phase GPU-event ms after the timed run")
    p.add_argument("--print-mem", action="store_true",
                   help="print peak device memory after the timed run")
    return p.parse_args()


def main():
    args = get_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world > 1

    cpu_fallback = not torch.cuda.is_available()
    if cpu_fallback:
        # debug path only; the driver always runs this on a GPU box
        args.batch_per_gpu = 8
        args.moco_k = 1024
        args.arch = "resnet18"
        args.crop_size = 32
        args.src_size = 40
        device = torch.device("cpu")
    else:
        device = torch.device("cuda", local_rank)
        torch.cuda.set_device(device)

    if distributed:
        dist.init_process_group(backend="gloo" if cpu_fallback else "nccl")

    torch.manual_seed(1234 + rank)
    model = moco_amd.MoCo(
        lambda num_classes: build_model(args.arch, num_classes=num_classes),
        dim=128, K=args.moco_k, m=0.999, T=0.2, mlp=True, shuffle_seed=1234)
    model = model.to(device)
    if not cpu_fallback:
        model = model.to(memory_format=torch.channels_last)
    if distributed:
        kw = dict(broadcast_buffers=False)
        if not cpu_fallback:
            kw["device_ids"] = [local_rank]
        model = torch.nn.parallel.DistributedDataParallel(model, **kw)

    optimizer = FusedSGD(model.parameters(), lr=0.03, momentum=0.9,
                         weight_decay=1e-4)
    stepper = MoCoStepper(model, optimizer, device, aug_plus=True,
                          out_size=args.crop_size,
                          dtype=torch.bfloat16 if not cpu_fallback
                          else torch.float32)

    # synthetic raw-image pool, resident on device (random data: the bf16
    # clock on zero-filled data flatters — guide §5.4 rule 25)
    g = torch.Generator(device="cpu").manual_seed(4321 + rank)
    pool = [torch.rand(args.batch_per_gpu, 3, args.src_size, args.src_size,
                       generator=g).to(device)
            for _ in range(args.pool_batches)]

    def one_step(i):
        stepper.step_from_raw(pool[i % len(pool)])

    for i in range(args.warmup):
        one_step(i)

    if args.phase_timing and not cpu_fallback:
        stepper.enable_timing(True)
        one_step(99)
        torch.cuda.synchronize()
        if rank == 0:
            print("phase_ms:", {k: round(v, 3)
                                for k, v in stepper.phase_times_ms().items()})
        stepper.enable_timing(False)

    if distributed:
        dist.barrier()
    if not cpu_fallback:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        one_step(args.warmup + i)
    if distributed:
        dist.barrier()
    if not cpu_fallback:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # MAX over ranks (slowest rank defines the job)
    if distributed:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device
                         if not cpu_fallback else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    n_gpus = world if distributed else 1
    total_images = args.batch_per_gpu * n_gpus * args.steps
    img_s = total_images / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if args.print_mem and not cpu_fallback and rank == 0:
        print("peak_mem_gb:",
              round(torch.cuda.max_memory_allocated() / 2**30, 2),
              "reserved_gb:",
              round(torch.cuda.max_memory_reserved() / 2**30, 2))

    if rank == 0:
        result = {
            "metric": "images/sec (whole node), ResNet-50 MoCo-v2 pretrain",
            "value": round(img_s, 2),
            "unit": "img/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": (round(img_s / BASELINE_NODE_IMG_S, 3)
                            if not cpu_fallback else None),
            "dtype": "bf16" if not cpu_fallback else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.arch,
                "moco": "v2 (mlp, T=0.2, aug+, K=%d)" % args.moco_k,
                "global_batch": args.batch_per_gpu * n_gpus,
                "crop": args.crop_size,
                "parallelism": f"dp{n_gpus}",
            },
        }
        print(json.dumps(result))

    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
