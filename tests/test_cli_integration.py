"""End-to-end CLI integration on CPU — BASELINE config 0: ResNet-18 MoCo-v1,
32x32 synthetic, queue=1024, world_size 1, no GPU."""

import os

import pytest
import subprocess
import sys

import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run(args, cwd=REPO, timeout=600):
    r = subprocess.run([sys.executable] + args, cwd=cwd, timeout=timeout,
                       capture_output=True, text=True)
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    return r.stdout


def test_config0_pretrain_then_lincls(tmp_path):
    out = run(["main_moco.py", "synthetic", "-a", "resnet18",
               "--moco-k", "1024", "--crop-size", "32", "--load-size", "40",
               "-b", "16", "-j", "0", "--epochs", "1",
               "--max-steps-per-epoch", "3", "--synthetic-len", "64",
               "--device", "cpu", "--save-dir", str(tmp_path),
               "-p", "1", "--seed", "1"])
    assert "Epoch: [0][2/3]" in out
    ckpt_path = tmp_path / "checkpoint_0000.pth.tar"
    assert ckpt_path.exists()
    ckpt = torch.load(str(ckpt_path), map_location="cpu", weights_only=False)
    assert ckpt["arch"] == "resnet18"
    assert ckpt["state_dict"]["module.queue"].shape == (128, 1024)

    out = run(["main_lincls.py", "synthetic", "-a", "resnet18",
               "--pretrained", str(ckpt_path), "--device", "cpu",
               "-b", "16", "-j", "0", "--epochs", "1",
               "--max-steps-per-epoch", "2", "--synthetic-len", "64",
               "--crop-size", "32", "--load-size", "40",
               "--num-classes", "10", "--save-dir", str(tmp_path),
               "-p", "1"])
    assert "sanity check passed" in out
    assert "Acc@1" in out


def test_resume(tmp_path):
    run(["main_moco.py", "synthetic", "-a", "resnet18", "--moco-k", "256",
         "--crop-size", "32", "--load-size", "40", "-b", "8", "-j", "0",
         "--epochs", "1", "--max-steps-per-epoch", "2",
         "--synthetic-len", "32", "--device", "cpu",
         "--save-dir", str(tmp_path), "-p", "1"])
    out = run(["main_moco.py", "synthetic", "-a", "resnet18",
               "--moco-k", "256", "--crop-size", "32", "--load-size", "40",
               "-b", "8", "-j", "0", "--epochs", "2",
               "--max-steps-per-epoch", "2", "--synthetic-len", "32",
               "--device", "cpu", "--save-dir", str(tmp_path), "-p", "1",
               "--resume", str(tmp_path / "checkpoint_0000.pth.tar")])
    assert "loaded checkpoint (epoch 1)" in out
    assert "Epoch: [1]" in out


def test_v2_flags(tmp_path):
    out = run(["main_moco.py", "synthetic", "-a", "resnet18",
               "--moco-k", "256", "--moco-t", "0.2", "--mlp", "--aug-plus",
               "--cos", "--crop-size", "32", "--load-size", "40",
               "-b", "8", "-j", "0", "--epochs", "1",
               "--max-steps-per-epoch", "2", "--synthetic-len", "32",
               "--device", "cpu", "--save-dir", str(tmp_path), "-p", "1"])
    assert "mlp=True" in out


def test_pretrain_on_imagefolder(tmp_path):
    """Real-image decode path end to end (PIL ImageFolder -> GPU-style
    batched aug -> MoCo step)."""
    pytest.importorskip("PIL")
    from PIL import Image
    for cls in ("a", "b"):
        d = tmp_path / "data" / "train" / cls
        d.mkdir(parents=True)
        for i in range(8):
            Image.new("RGB", (50, 70), color=(i * 20, 50, 100)).save(
                str(d / f"{i}.png"))
    out = run(["main_moco.py", str(tmp_path / "data"), "-a", "resnet18",
               "--moco-k", "32", "--crop-size", "32", "--load-size", "40",
               "-b", "8", "-j", "0", "--epochs", "1",
               "--max-steps-per-epoch", "2", "--device", "cpu",
               "--save-dir", str(tmp_path), "-p", "1"])
    assert "Epoch: [0][1/2]" in out


def test_lincls_evaluate_only(tmp_path):
    run(["main_moco.py", "synthetic", "-a", "resnet18", "--moco-k", "32",
         "--crop-size", "32", "--load-size", "40", "-b", "8", "-j", "0",
         "--epochs", "1", "--max-steps-per-epoch", "1",
         "--synthetic-len", "16", "--device", "cpu",
         "--save-dir", str(tmp_path), "-p", "1"])
    out = run(["main_lincls.py", "synthetic", "-a", "resnet18",
               "--pretrained", str(tmp_path / "checkpoint_0000.pth.tar"),
               "--device", "cpu", "-b", "8", "-j", "0", "--evaluate",
               "--synthetic-len", "32", "--crop-size", "32",
               "--load-size", "40", "--num-classes", "4",
               "--save-dir", str(tmp_path)])
    assert "Acc@1" in out


def test_mp_spawn_launch_path(tmp_path):
    """The reference-style mp.spawn launcher (--multiprocessing-distributed)
    on one CPU 'device'."""
    out = run(["main_moco.py", "synthetic", "-a", "resnet18",
               "--moco-k", "32", "--crop-size", "32", "--load-size", "40",
               "-b", "8", "-j", "0", "--epochs", "1",
               "--max-steps-per-epoch", "1", "--synthetic-len", "16",
               "--device", "cpu", "--save-dir", str(tmp_path), "-p", "1",
               "--multiprocessing-distributed", "--dist-backend", "gloo",
               "--dist-url", "tcp://127.0.0.1:29411"])
    assert "Epoch: [0]" in out
