#!/usr/bin/env python
"""MoCo pretraining CLI — MI355X-native engine.

Flag surface is a superset of the reference main_moco.py (argparse inventory
at reference main_moco.py:38-101; launch skeleton :103-134, worker :137-287).
MoCo v2 = --mlp --moco-t 0.2 --aug-plus --cos.

Extra flags vs the reference: --data synthetic (no-network environments),
--device {cuda,cpu}, --moco-dtype {fp32,bf16}, --synthetic-len, --load-size.
"""

import argparse
import builtins
import os
import random
import warnings

import torch
import torch.distributed as dist
import torch.multiprocessing as mp
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

import moco_amd
from moco_amd.engine import MoCoStepper, train_epoch
from moco_amd.loader import build_dataset
from moco_amd.models.resnet import MODEL_REGISTRY, build_model
from moco_amd.ops import FusedSGD
from moco_amd.utils import adjust_learning_rate, save_checkpoint

model_names = sorted(MODEL_REGISTRY)


def get_parser():
    parser = argparse.ArgumentParser(description="MoCo pretraining (MI355X)")
    parser.add_argument("data", metavar="DIR", nargs="?", default="synthetic",
                        help="path to dataset, or 'synthetic'")
    parser.add_argument("-a", "--arch", default="resnet50", choices=model_names)
    parser.add_argument("-j", "--workers", default=32, type=int)
    parser.add_argument("--epochs", default=200, type=int)
    parser.add_argument("--start-epoch", default=0, type=int)
    parser.add_argument("-b", "--batch-size", default=256, type=int,
                        help="total batch size over all GPUs")
    parser.add_argument("--lr", "--learning-rate", default=0.03, type=float,
                        dest="lr")
    parser.add_argument("--schedule", default=[120, 160], nargs="*", type=int)
    parser.add_argument("--momentum", default=0.9, type=float)
    parser.add_argument("--wd", "--weight-decay", default=1e-4, type=float,
                        dest="weight_decay")
    parser.add_argument("-p", "--print-freq", default=10, type=int)
    parser.add_argument("--resume", default="", type=str)
    parser.add_argument("--world-size", default=-1, type=int)
    parser.add_argument("--rank", default=-1, type=int)
    parser.add_argument("--dist-url", default="tcp://127.0.0.1:23456",
                        type=str)
    parser.add_argument("--dist-backend", default="nccl", type=str)
    parser.add_argument("--seed", default=None, type=int)
    parser.add_argument("--gpu", default=None, type=int)
    parser.add_argument("--multiprocessing-distributed", action="store_true")
    # MoCo specific
    parser.add_argument("--moco-dim", default=128, type=int)
    parser.add_argument("--moco-k", default=65536, type=int)
    parser.add_argument("--moco-m", default=0.999, type=float)
    parser.add_argument("--moco-t", default=0.07, type=float)
    # MoCo v2
    parser.add_argument("--mlp", action="store_true")
    parser.add_argument("--aug-plus", action="store_true")
    parser.add_argument("--cos", action="store_true")
    # MI355X engine knobs
    parser.add_argument("--device", default=None, choices=["cuda", "cpu"],
                        help="default: cuda if available else cpu")
    parser.add_argument("--moco-dtype", default="bf16",
                        choices=["fp32", "bf16"])
    parser.add_argument("--crop-size", default=224, type=int)
    parser.add_argument("--load-size", default=256, type=int)
    parser.add_argument("--synthetic-len", default=2048, type=int)
    parser.add_argument("--save-dir", default=".", type=str)
    parser.add_argument("--max-steps-per-epoch", default=0, type=int,
                        help="truncate epochs (smoke/testing); 0 = full")
    parser.add_argument("--dist-timeout", default=1800, type=int,
                        help="collective timeout (s); a wedged rank aborts "
                             "the job instead of hanging (SURVEY §5.3)")
    parser.add_argument("--deterministic", action="store_true",
                        help="deterministic kernels + no MIOpen benchmark "
                             "(debug/tests; slower)")
    parser.add_argument("--metrics-file", default="metrics.jsonl", type=str,
                        help="JSONL metrics stream (rank 0; '' disables)")
    parser.add_argument("--debug-serialize", action="store_true",
                        help="serialize+sync every kernel launch "
                             "(AMD_SERIALIZE_KERNEL=3, HIP_LAUNCH_BLOCKING) "
                             "— the ROCm race-hunting mode; very slow")
    return parser


def main(argv=None):
    args = get_parser().parse_args(argv)

    if args.debug_serialize:
        # must be set before the first HIP call in every worker
        os.environ["AMD_SERIALIZE_KERNEL"] = "3"
        os.environ["HIP_LAUNCH_BLOCKING"] = "1"

    if args.seed is not None:
        random.seed(args.seed)
        torch.manual_seed(args.seed)
        warnings.warn("You have chosen to seed training.")

    if args.device is None:
        args.device = "cuda" if torch.cuda.is_available() else "cpu"

    ngpus_per_node = torch.cuda.device_count() if args.device == "cuda" else 1
    if args.multiprocessing_distributed:
        args.world_size = ngpus_per_node * max(1, args.world_size)
        mp.spawn(main_worker, nprocs=ngpus_per_node,
                 args=(ngpus_per_node, args))
    else:
        # torchrun / env:// style, or single process
        if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
            args.rank = int(os.environ["RANK"])
            args.world_size = int(os.environ["WORLD_SIZE"])
            args.dist_url = "env://"
            gpu = int(os.environ.get("LOCAL_RANK", 0))
        else:
            gpu = args.gpu
        main_worker(gpu, ngpus_per_node, args)


def main_worker(gpu, ngpus_per_node, args):
    args.gpu = gpu
    distributed = args.world_size > 1 or (args.multiprocessing_distributed
                                          and ngpus_per_node > 1)
    if args.deterministic:
        torch.use_deterministic_algorithms(True, warn_only=True)
        torch.backends.cudnn.deterministic = True
        torch.backends.cudnn.benchmark = False
    if distributed:
        # fail fast on a wedged rank instead of hanging RCCL
        os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "1")
        if args.multiprocessing_distributed:
            args.rank = args.rank * ngpus_per_node + gpu if args.rank > 0 \
                else gpu
        from datetime import timedelta
        dist.init_process_group(backend=args.dist_backend,
                                init_method=args.dist_url,
                                world_size=args.world_size, rank=args.rank,
                                timeout=timedelta(seconds=args.dist_timeout))
        if args.rank != 0:  # silence non-master ranks (ref main_moco.py:139)
            def print_pass(*a, **k):
                pass
            builtins.print = print_pass

    device = torch.device(args.device,
                          gpu if (args.device == "cuda" and gpu is not None)
                          else 0) if args.device == "cuda" else \
        torch.device("cpu")
    if args.device == "cuda":
        torch.cuda.set_device(device)

    print(f"=> creating model MoCo({args.arch}, dim={args.moco_dim}, "
          f"K={args.moco_k}, m={args.moco_m}, T={args.moco_t}, "
          f"mlp={args.mlp})")
    seed = args.seed if args.seed is not None else 0
    model = moco_amd.MoCo(
        lambda num_classes: build_model(args.arch, num_classes=num_classes),
        dim=args.moco_dim, K=args.moco_k, m=args.moco_m, T=args.moco_t,
        mlp=args.mlp, shuffle_seed=seed)
    model = model.to(device)
    if args.device == "cuda":
        model = model.to(memory_format=torch.channels_last)

    per_rank_batch = args.batch_size
    if distributed:
        per_rank_batch = args.batch_size // dist.get_world_size()
        ddp_kwargs = dict(broadcast_buffers=False)
        if args.device == "cuda":
            ddp_kwargs["device_ids"] = [device.index]
        model = torch.nn.parallel.DistributedDataParallel(model, **ddp_kwargs)

    optimizer = FusedSGD(model.parameters(), args.lr,
                         momentum=args.momentum,
                         weight_decay=args.weight_decay)

    if args.resume:
        if os.path.isfile(args.resume):
            print(f"=> loading checkpoint '{args.resume}'")
            ckpt = torch.load(args.resume, map_location="cpu",
                              weights_only=False)
            args.start_epoch = ckpt["epoch"]
            (model.module if distributed else model).load_state_dict(
                {k[len("module."):] if k.startswith("module.") else k: v
                 for k, v in ckpt["state_dict"].items()})
            optimizer.load_state_dict(ckpt["optimizer"])
            print(f"=> loaded checkpoint (epoch {ckpt['epoch']})")
        else:
            print(f"=> no checkpoint found at '{args.resume}'")

    dataset = build_dataset(args.data, load_size=args.load_size,
                            synthetic_len=args.synthetic_len,
                            seed=seed)
    sampler = DistributedSampler(dataset) if distributed else None
    loader = DataLoader(dataset, batch_size=per_rank_batch,
                        shuffle=(sampler is None), sampler=sampler,
                        num_workers=args.workers, pin_memory=True,
                        drop_last=True)

    dtype = torch.bfloat16 if args.moco_dtype == "bf16" else torch.float32
    stepper = MoCoStepper(model, optimizer, device,
                          aug_plus=args.aug_plus, out_size=args.crop_size,
                          dtype=dtype)
    if args.deterministic:
        torch.backends.cudnn.benchmark = False  # stepper turns it on

    from moco_amd.utils import MetricsLogger
    is_master = not distributed or dist.get_rank() == 0
    metrics = MetricsLogger(
        os.path.join(args.save_dir, args.metrics_file)
        if (is_master and args.metrics_file) else None)

    for epoch in range(args.start_epoch, args.epochs):
        if sampler is not None:
            sampler.set_epoch(epoch)
        adjust_learning_rate(optimizer, epoch, args)

        if args.max_steps_per_epoch:
            from itertools import islice
            it = islice(iter(loader), args.max_steps_per_epoch)

            class _L:
                def __iter__(self):
                    return it

                def __len__(self):
                    return args.max_steps_per_epoch
            train_epoch(_L(), stepper, epoch, args, metrics=metrics)
        else:
            train_epoch(loader, stepper, epoch, args, metrics=metrics)

        if distributed:
            # DDP runs with broadcast_buffers=False; average BN running
            # stats across ranks so the checkpoint reflects the whole job
            from moco_amd.parallel import sync_bn_running_stats
            sync_bn_running_stats(model.module)
        if not distributed or dist.get_rank() == 0:
            raw_model = model.module if distributed else model
            # checkpoint contract: DDP-prefixed keys (module.encoder_q.* ...)
            state_dict = {"module." + k: v
                          for k, v in raw_model.state_dict().items()}
            save_checkpoint(
                {"epoch": epoch + 1, "arch": args.arch,
                 "state_dict": state_dict,
                 "optimizer": optimizer.state_dict()},
                is_best=False,
                filename=os.path.join(
                    args.save_dir, f"checkpoint_{epoch:04d}.pth.tar"))


if __name__ == "__main__":
    main()
