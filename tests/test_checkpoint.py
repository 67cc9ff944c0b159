import os
import subprocess
import sys

import torch

import moco_amd
from moco_amd.models.resnet import resnet18
from moco_amd.ops import FusedSGD
from moco_amd.utils.checkpoint import (load_pretrained_backbone,
                                       sanity_check, strip_moco_prefix)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def make_ckpt(tmp_path, mlp=False):
    torch.manual_seed(0)
    model = moco_amd.MoCo(lambda num_classes: resnet18(num_classes=num_classes),
                          dim=128, K=64, mlp=mlp)
    opt = FusedSGD(model.parameters(), lr=0.03, momentum=0.9)
    logits, _ = model(torch.randn(4, 3, 32, 32), torch.randn(4, 3, 32, 32))
    loss = moco_amd.moco_loss(logits)
    loss.backward()
    opt.step()
    path = str(tmp_path / "checkpoint_0000.pth.tar")
    state = {"epoch": 1, "arch": "resnet18",
             "state_dict": {"module." + k: v
                            for k, v in model.state_dict().items()},
             "optimizer": opt.state_dict()}
    torch.save(state, path)
    return model, opt, path


def test_checkpoint_key_contract(tmp_path):
    model, _, path = make_ckpt(tmp_path)
    ckpt = torch.load(path, map_location="cpu", weights_only=False)
    keys = set(ckpt["state_dict"].keys())
    assert "module.queue" in keys and "module.queue_ptr" in keys
    assert any(k.startswith("module.encoder_q.conv1") for k in keys)
    assert any(k.startswith("module.encoder_k.layer4") for k in keys)
    assert ckpt["state_dict"]["module.queue"].shape == (128, 64)
    assert ckpt["state_dict"]["module.queue_ptr"].shape == (1,)
    assert set(ckpt) == {"epoch", "arch", "state_dict", "optimizer"}


def test_resume_roundtrip(tmp_path):
    model, opt, path = make_ckpt(tmp_path)
    ckpt = torch.load(path, map_location="cpu", weights_only=False)
    torch.manual_seed(99)
    model2 = moco_amd.MoCo(
        lambda num_classes: resnet18(num_classes=num_classes), dim=128, K=64)
    model2.load_state_dict(
        {k[len("module."):]: v for k, v in ckpt["state_dict"].items()})
    for (k1, v1), (k2, v2) in zip(model.state_dict().items(),
                                  model2.state_dict().items()):
        assert k1 == k2 and torch.equal(v1, v2), k1
    opt2 = FusedSGD(model2.parameters(), lr=0.03, momentum=0.9)
    opt2.load_state_dict(ckpt["optimizer"])  # must not raise


def test_lincls_load_and_sanity(tmp_path):
    _, _, path = make_ckpt(tmp_path)
    clf = resnet18(num_classes=1000)
    load_pretrained_backbone(clf, path)
    sanity_check(clf.state_dict(), path)  # untouched backbone passes
    # mutate a backbone weight -> sanity check must fail
    clf.layer1[0].conv1.weight.data.add_(1.0)
    try:
        sanity_check(clf.state_dict(), path)
        raise RuntimeError("sanity_check should have failed")
    except AssertionError:
        pass


def test_lincls_load_mlp_head_ckpt(tmp_path):
    """v2 checkpoints (MLP head) load the same way: fc.* keys are dropped."""
    _, _, path = make_ckpt(tmp_path, mlp=True)
    clf = resnet18(num_classes=1000)
    load_pretrained_backbone(clf, path)


def test_detectron2_converter(tmp_path):
    import pickle
    _, _, path = make_ckpt(tmp_path)
    out = str(tmp_path / "d2.pkl")
    r = subprocess.run(
        [sys.executable,
         os.path.join(REPO, "detection", "convert-pretrain-to-detectron2.py"),
         path, out],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    with open(out, "rb") as f:
        blob = pickle.load(f)
    assert blob["__author__"] == "MOCO"
    assert blob["matching_heuristics"] is True
    keys = set(blob["model"].keys())
    assert "stem.conv1.weight" in keys
    assert any(k.startswith("res2.") for k in keys)
    assert any(".shortcut." in k or ".shortcut" in k for k in keys)
    assert not any(k.startswith("fc.") for k in keys)


def test_resume_matches_continuous_run(tmp_path):
    """Save at step 2, reload into a fresh model, continue with the same data
    and step ids: the final state must match the uninterrupted 4-step run
    (checkpoint completeness: params, EMA state, queue, ptr, momentum)."""
    import copy

    def make(seed=5):
        torch.manual_seed(seed)
        m = moco_amd.MoCo(lambda num_classes: resnet18(num_classes=num_classes),
                          dim=128, K=64, m=0.99, T=0.2, shuffle_seed=3)
        o = FusedSGD(m.parameters(), lr=0.01, momentum=0.9, weight_decay=1e-4)
        return m, o

    torch.manual_seed(77)
    data = [(torch.randn(4, 3, 32, 32), torch.randn(4, 3, 32, 32))
            for _ in range(4)]

    def run_steps(model, opt, steps):
        for s in steps:
            logits, _ = model(data[s][0], data[s][1], step=s)
            loss = moco_amd.moco_loss(logits)
            opt.zero_grad()
            loss.backward()
            opt.step()

    # continuous 4-step run
    m1, o1 = make()
    run_steps(m1, o1, [0, 1, 2, 3])

    # interrupted run: 2 steps -> checkpoint -> fresh objects -> 2 steps
    m2, o2 = make()
    run_steps(m2, o2, [0, 1])
    state = {"state_dict": copy.deepcopy(m2.state_dict()),
             "optimizer": copy.deepcopy(o2.state_dict())}
    m3, o3 = make(seed=99)  # different init: everything must come from ckpt
    m3.load_state_dict(state["state_dict"])
    o3 = FusedSGD(m3.parameters(), lr=0.01, momentum=0.9, weight_decay=1e-4)
    o3.load_state_dict(state["optimizer"])
    run_steps(m3, o3, [2, 3])

    sd1, sd3 = m1.state_dict(), m3.state_dict()
    for k in sd1:
        if "num_batches_tracked" in k:
            assert torch.equal(sd1[k], sd3[k]), k
        else:
            assert torch.allclose(sd1[k], sd3[k], atol=1e-5), \
                (k, (sd1[k] - sd3[k]).abs().max())
