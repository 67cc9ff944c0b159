import json
import math

import torch

from moco_amd.utils import (AverageMeter, MetricsLogger, ProgressMeter,
                            accuracy, adjust_learning_rate)


class _Args:
    def __init__(self, lr=0.03, schedule=(120, 160), cos=False, epochs=200):
        self.lr = lr
        self.schedule = list(schedule)
        self.cos = cos
        self.epochs = epochs


def test_average_meter():
    m = AverageMeter("x", ":.2f")
    m.update(1.0, n=2)
    m.update(4.0, n=1)
    assert m.count == 3 and abs(m.avg - 2.0) < 1e-9
    assert "x" in str(m)


def test_progress_meter_runs(capsys):
    m = AverageMeter("loss", ":.2f")
    p = ProgressMeter(100, [m], prefix="Epoch: [3]")
    p.display(7)
    out = capsys.readouterr().out
    assert "Epoch: [3]" in out and "loss" in out


def test_accuracy_topk():
    logits = torch.tensor([[0.9, 0.1, 0.0], [0.1, 0.8, 0.1],
                           [0.3, 0.3, 0.4], [1.0, 0.0, 0.0]])
    target = torch.tensor([0, 1, 0, 2])
    acc1, acc2 = accuracy(logits, target, topk=(1, 2))
    assert abs(acc1.item() - 50.0) < 1e-5   # rows 0,1 correct
    assert abs(acc2.item() - 75.0) < 1e-5   # row 2's class 0 is 2nd


def test_step_lr_schedule():
    opt = torch.optim.SGD([torch.nn.Parameter(torch.zeros(1))], lr=0.03)
    args = _Args()
    assert abs(adjust_learning_rate(opt, 0, args) - 0.03) < 1e-9
    assert abs(adjust_learning_rate(opt, 120, args) - 0.003) < 1e-9
    assert abs(adjust_learning_rate(opt, 160, args) - 0.0003) < 1e-9
    assert opt.param_groups[0]["lr"] == adjust_learning_rate(opt, 160, args)


def test_cos_lr_schedule():
    opt = torch.optim.SGD([torch.nn.Parameter(torch.zeros(1))], lr=0.03)
    args = _Args(cos=True)
    assert abs(adjust_learning_rate(opt, 0, args) - 0.03) < 1e-9
    mid = adjust_learning_rate(opt, 100, args)
    assert abs(mid - 0.015) < 1e-6
    assert adjust_learning_rate(opt, 199, args) < 1e-5 * 0.03 * 1e3


def test_metrics_logger(tmp_path):
    path = str(tmp_path / "m.jsonl")
    ml = MetricsLogger(path)
    ml.log(epoch=0, loss=1.5)
    ml.log(epoch=0, step=1, loss=1.2)
    ml.close()
    lines = [json.loads(l) for l in open(path)]
    assert len(lines) == 2
    assert lines[0]["loss"] == 1.5 and "ts" in lines[0]
    # disabled logger is a no-op
    ml2 = MetricsLogger(None)
    ml2.log(x=1)
    ml2.close()


def test_loader_synthetic_determinism():
    from moco_amd.loader import SyntheticImageDataset
    ds = SyntheticImageDataset(length=16, size=8, seed=7)
    a, la = ds[3]
    b, lb = ds[3]
    c, _ = ds[4]
    assert torch.equal(a, b) and la == lb
    assert not torch.equal(a, c)
    assert a.shape == (3, 8, 8) and 0.0 <= a.min() and a.max() <= 1.0
