"""LR schedule — parity with reference adjust_learning_rate
(main_moco.py:395-403): step decay x0.1 at --schedule epochs, or cosine with
--cos."""

import math


def adjust_learning_rate(optimizer, epoch: int, args) -> float:
    lr = args.lr
    if getattr(args, "cos", False):
        lr *= 0.5 * (1.0 + math.cos(math.pi * epoch / args.epochs))
    else:
        for milestone in args.schedule:
            lr *= 0.1 if epoch >= milestone else 1.0
    for param_group in optimizer.param_groups:
        param_group["lr"] = lr
    return lr
