#!/usr/bin/env python
"""Linear evaluation (frozen-backbone linear probe) — MI355X-native engine.

Parity with the reference main_lincls.py: freeze everything but fc
(≈:160-170), load module.encoder_q.* from the pretrain checkpoint stripping
the prefix and assert only fc.{weight,bias} are missing (≈:172-190), optimize
exactly 2 tensors (≈:200), lr 30 / schedule [60,80] / wd 0 defaults
(≈:40-95), model.eval() during training so frozen BN uses running stats
(≈:300), top-1/5 validation + best-checkpoint tracking, and the bitwise
sanity_check (≈:400-425).
"""

import argparse
import builtins
import os
import random
import time
import warnings

import torch
import torch.distributed as dist
import torch.nn as nn
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

from moco_amd.loader import build_dataset, cpu_two_crop_eval_transform
from moco_amd.models.resnet import MODEL_REGISTRY, build_model
from moco_amd.ops.augment import (sample_two_crop_params,
                                  single_crop_augment)
from moco_amd.utils import (AverageMeter, ProgressMeter, accuracy,
                            adjust_learning_rate, load_pretrained_backbone,
                            sanity_check, save_checkpoint)

model_names = sorted(MODEL_REGISTRY)


def get_parser():
    p = argparse.ArgumentParser(description="Linear evaluation (MI355X)")
    p.add_argument("data", metavar="DIR", nargs="?", default="synthetic")
    p.add_argument("-a", "--arch", default="resnet50", choices=model_names)
    p.add_argument("-j", "--workers", default=32, type=int)
    p.add_argument("--epochs", default=100, type=int)
    p.add_argument("--start-epoch", default=0, type=int)
    p.add_argument("-b", "--batch-size", default=256, type=int)
    p.add_argument("--lr", "--learning-rate", default=30.0, type=float,
                   dest="lr")
    p.add_argument("--schedule", default=[60, 80], nargs="*", type=int)
    p.add_argument("--momentum", default=0.9, type=float)
    p.add_argument("--wd", "--weight-decay", default=0.0, type=float,
                   dest="weight_decay")
    p.add_argument("-p", "--print-freq", default=10, type=int)
    p.add_argument("--resume", default="", type=str)
    p.add_argument("--world-size", default=-1, type=int)
    p.add_argument("--rank", default=-1, type=int)
    p.add_argument("--dist-url", default="tcp://127.0.0.1:23457", type=str)
    p.add_argument("--dist-backend", default="nccl", type=str)
    p.add_argument("--seed", default=None, type=int)
    p.add_argument("--gpu", default=None, type=int)
    p.add_argument("--multiprocessing-distributed", action="store_true")
    p.add_argument("--pretrained", default="", type=str,
                   help="path to moco pretrained checkpoint")
    p.add_argument("-e", "--evaluate", action="store_true",
                   help="evaluate on the validation set and exit")
    p.add_argument("--device", default=None, choices=["cuda", "cpu"])
    p.add_argument("--lincls-dtype", default="fp32", choices=["fp32", "bf16"],
                   help="bf16 autocast for the frozen-backbone forward")
    p.add_argument("--crop-size", default=224, type=int)
    p.add_argument("--load-size", default=256, type=int)
    p.add_argument("--synthetic-len", default=1024, type=int)
    p.add_argument("--num-classes", default=1000, type=int)
    p.add_argument("--save-dir", default=".", type=str)
    p.add_argument("--max-steps-per-epoch", default=0, type=int)
    return p


def train_transform_batch(raw: torch.Tensor, crop: int,
                          dtype=None, channels_last=False):
    """RandomResizedCrop + flip + normalize on a raw batch (no jitter/blur —
    reference lincls train aug).  Runs on whatever device `raw` lives on
    (the HIP augmentation kernels when on GPU)."""
    B, _, H, W = raw.shape
    # torchvision-default crop scale (0.08, 1.0) — the reference lincls uses
    # plain RandomResizedCrop, not MoCo's (0.2, 1.0) pretrain scale
    params = sample_two_crop_params(B, H, W, crop, aug_plus=False,
                                    scale=(0.08, 1.0))
    params = params[:B]
    # disable jitter/grayscale (lincls aug has none)
    params[:, 5] = 0.0   # gray
    params[:, 7:10] = 1.0  # brightness/contrast/saturation factors
    params[:, 10] = 0.0  # hue
    params[:, 6] = 0.0   # sigma
    return single_crop_augment(raw, crop, params.to(raw.device),
                               out_dtype=dtype, channels_last=channels_last)


def main(argv=None):
    args = get_parser().parse_args(argv)
    if args.seed is not None:
        random.seed(args.seed)
        torch.manual_seed(args.seed)
        warnings.warn("You have chosen to seed training.")
    if args.device is None:
        args.device = "cuda" if torch.cuda.is_available() else "cpu"

    distributed = args.world_size > 1
    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        args.rank = int(os.environ["RANK"])
        args.world_size = int(os.environ["WORLD_SIZE"])
        distributed = args.world_size > 1
        if distributed:
            dist.init_process_group(backend=args.dist_backend,
                                    init_method="env://")
            if args.rank != 0:
                builtins.print = lambda *a, **k: None

    device = torch.device(args.device)
    if args.device == "cuda":
        local = int(os.environ.get("LOCAL_RANK", args.gpu or 0))
        device = torch.device("cuda", local)
        torch.cuda.set_device(device)

    print(f"=> creating model '{args.arch}'")
    model = build_model(args.arch, num_classes=args.num_classes)

    # freeze all layers but the last fc
    for name, param in model.named_parameters():
        if name not in ("fc.weight", "fc.bias"):
            param.requires_grad = False
    model.fc.weight.data.normal_(mean=0.0, std=0.01)
    model.fc.bias.data.zero_()

    if args.pretrained:
        load_pretrained_backbone(model, args.pretrained)
        print(f"=> loaded pre-trained model '{args.pretrained}'")

    model = model.to(device)
    if args.device == "cuda":
        model = model.to(memory_format=torch.channels_last)
    if distributed:
        kw = dict(broadcast_buffers=False)
        if args.device == "cuda":
            kw["device_ids"] = [device.index]
        model = torch.nn.parallel.DistributedDataParallel(model, **kw)

    criterion = nn.CrossEntropyLoss().to(device)
    parameters = [p for p in model.parameters() if p.requires_grad]
    assert len(parameters) == 2  # fc.weight, fc.bias (ref main_lincls ≈:200)
    optimizer = torch.optim.SGD(parameters, args.lr, momentum=args.momentum,
                                weight_decay=args.weight_decay)

    best_acc1 = 0.0
    if args.resume and os.path.isfile(args.resume):
        ckpt = torch.load(args.resume, map_location="cpu", weights_only=False)
        args.start_epoch = ckpt["epoch"]
        best_acc1 = ckpt.get("best_acc1", 0.0)
        (model.module if distributed else model).load_state_dict(
            {k[len("module."):] if k.startswith("module.") else k: v
             for k, v in ckpt["state_dict"].items()})
        optimizer.load_state_dict(ckpt["optimizer"])
        print(f"=> resumed from '{args.resume}' (epoch {ckpt['epoch']})")

    train_set = build_dataset(args.data, load_size=args.load_size,
                              synthetic_len=args.synthetic_len,
                              seed=args.seed or 0,
                              num_classes=args.num_classes)
    val_set = build_dataset(args.data, load_size=args.load_size,
                            synthetic_len=max(64, args.synthetic_len // 4),
                            seed=(args.seed or 0) + 999,
                            num_classes=args.num_classes)
    train_sampler = DistributedSampler(train_set) if distributed else None
    train_loader = DataLoader(train_set, batch_size=args.batch_size,
                              shuffle=(train_sampler is None),
                              sampler=train_sampler,
                              num_workers=args.workers, drop_last=True)
    val_loader = DataLoader(val_set, batch_size=args.batch_size,
                            shuffle=False, num_workers=args.workers)

    if args.evaluate:
        return validate(val_loader, model, criterion, device, args)

    for epoch in range(args.start_epoch, args.epochs):
        if train_sampler is not None:
            train_sampler.set_epoch(epoch)
        adjust_learning_rate(optimizer, epoch, args)
        train(train_loader, model, criterion, optimizer, epoch, device, args)
        acc1 = validate(val_loader, model, criterion, device, args)

        is_best = acc1 > best_acc1
        best_acc1 = max(acc1, best_acc1)
        if not distributed or dist.get_rank() == 0:
            raw_model = model.module if distributed else model
            save_checkpoint(
                {"epoch": epoch + 1, "arch": args.arch,
                 "state_dict": {"module." + k: v
                                for k, v in raw_model.state_dict().items()},
                 "best_acc1": best_acc1,
                 "optimizer": optimizer.state_dict()},
                is_best,
                filename=os.path.join(args.save_dir, "checkpoint.pth.tar"))
            if epoch == args.start_epoch and args.pretrained:
                sanity_check(raw_model.state_dict(), args.pretrained)
    return best_acc1


def train(loader, model, criterion, optimizer, epoch, device, args):
    losses = AverageMeter("Loss", ":.4e")
    top1 = AverageMeter("Acc@1", ":6.2f")
    top5 = AverageMeter("Acc@5", ":6.2f")
    progress = ProgressMeter(len(loader), [losses, top1, top5],
                             prefix=f"Epoch: [{epoch}]")
    # eval mode on purpose: frozen BN uses running stats (ref ≈:300)
    model.eval()
    use_bf16 = args.lincls_dtype == "bf16" and device.type == "cuda"
    for i, (raw, target) in enumerate(loader):
        if args.max_steps_per_epoch and i >= args.max_steps_per_epoch:
            break
        images = train_transform_batch(
            raw.to(device, non_blocking=True), args.crop_size,
            dtype=torch.bfloat16 if use_bf16 else None,
            channels_last=(device.type == "cuda"))
        target = target.to(device)
        with torch.autocast("cuda", dtype=torch.bfloat16, enabled=use_bf16):
            output = model(images)
        loss = criterion(output.float(), target)
        acc1, acc5 = accuracy(output, target, topk=(1, 5))
        losses.update(loss.item(), images.size(0))
        top1.update(acc1[0].item(), images.size(0))
        top5.update(acc5[0].item(), images.size(0))
        optimizer.zero_grad()
        loss.backward()
        optimizer.step()
        if i % args.print_freq == 0:
            progress.display(i)


@torch.no_grad()
def validate(loader, model, criterion, device, args):
    losses = AverageMeter("Loss", ":.4e")
    top1 = AverageMeter("Acc@1", ":6.2f")
    top5 = AverageMeter("Acc@5", ":6.2f")
    model.eval()
    tfm = cpu_two_crop_eval_transform(args.crop_size)
    for i, (raw, target) in enumerate(loader):
        if args.max_steps_per_epoch and i >= args.max_steps_per_epoch:
            break
        images = tfm(raw).to(device)  # batched resize/crop/normalize
        target = target.to(device)
        output = model(images)
        loss = criterion(output, target)
        losses.update(loss.item(), images.size(0))
        acc1, acc5 = accuracy(output, target, topk=(1, 5))
        top1.update(acc1[0].item(), images.size(0))
        top5.update(acc5[0].item(), images.size(0))
    print(f" * Acc@1 {top1.avg:.3f} Acc@5 {top5.avg:.3f} "
          f"Loss {losses.avg:.4f}")
    return top1.avg


if __name__ == "__main__":
    main()
