"""a distributed run must match the single-process run at the
same global batch when BN partitioning is taken out of the picture (eval
mode -> running stats): per-step logits and the queue agree exactly."""

import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import moco_amd
from moco_amd.models.resnet import resnet18


def _build(seed=11):
    torch.manual_seed(seed)
    m = moco_amd.MoCo(lambda num_classes: resnet18(num_classes=num_classes),
                      dim=128, K=32, m=0.99, T=0.2, shuffle_seed=9)
    m.eval()
    return m


def _data():
    g = torch.Generator().manual_seed(123)
    return [(torch.randn(8, 3, 32, 32, generator=g),
             torch.randn(8, 3, 32, 32, generator=g)) for _ in range(2)]


def _worker(rank, world, port, outdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    model = _build()
    data = _data()
    n = 8 // world
    sl = slice(rank * n, (rank + 1) * n)
    outs = []
    with torch.no_grad():
        for step, (im_q, im_k) in enumerate(data):
            logits, _ = model(im_q[sl], im_k[sl], step=step)
            outs.append(logits)
    torch.save({"logits": torch.cat(outs), "queue": model.queue,
                "ptr": model.queue_ptr}, f"{outdir}/rank{rank}.pt")
    dist.destroy_process_group()


def test_world2_matches_world1(tmp_path):
    mp.spawn(_worker, args=(2, 29420, str(tmp_path)), nprocs=2, join=True)

    # single-process reference on the full global batch
    model = _build()
    data = _data()
    ref_logits = []
    with torch.no_grad():
        for step, (im_q, im_k) in enumerate(data):
            logits, _ = model(im_q, im_k, step=step)
            ref_logits.append(logits)

    r0 = torch.load(str(tmp_path / "rank0.pt"), weights_only=False)
    r1 = torch.load(str(tmp_path / "rank1.pt"), weights_only=False)
    assert torch.allclose(r0["queue"], r1["queue"], atol=1e-6)
    assert torch.allclose(r0["queue"], model.queue, atol=1e-5)
    assert r0["ptr"].item() == model.queue_ptr.item() == (2 * 8) % 32
    # logits: rank r holds rows [r*4, r*4+4) of each step
    for step in range(2):
        ref = ref_logits[step]
        got = torch.cat([r0["logits"][step * 4:(step + 1) * 4],
                         r1["logits"][step * 4:(step + 1) * 4]])
        assert torch.allclose(got, ref, atol=1e-4), \
            (step, (got - ref).abs().max())
