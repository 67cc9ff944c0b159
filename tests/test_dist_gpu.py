"""RCCL-path distributed tests on a single GPU (world_size 1): exercises
the real nccl backend, DDP wrapping, the all_to_all shuffle exchange and the
comm-stream overlap machinery end to end."""

import os

import pytest
import torch
import torch.distributed as dist

pytestmark = pytest.mark.gpu


@pytest.fixture
def nccl_world1():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29512")
    dist.init_process_group("nccl", rank=0, world_size=1)
    yield
    dist.destroy_process_group()


def test_ddp_train_step_nccl(nccl_world1):
    import moco_amd
    from moco_amd.engine import MoCoStepper
    from moco_amd.models.resnet import resnet18
    from moco_amd.ops import FusedSGD

    torch.manual_seed(0)
    device = torch.device("cuda", 0)
    model = moco_amd.MoCo(
        lambda num_classes: resnet18(num_classes=num_classes),
        dim=128, K=512, m=0.999, T=0.2, mlp=True)
    model = model.to(device).to(memory_format=torch.channels_last)
    model = torch.nn.parallel.DistributedDataParallel(
        model, device_ids=[0], broadcast_buffers=False)
    opt = FusedSGD(model.parameters(), lr=0.03, momentum=0.9,
                   weight_decay=1e-4)
    stepper = MoCoStepper(model, opt, device, aug_plus=True, out_size=64)
    for _ in range(2):
        raw = torch.rand(8, 3, 96, 96, device=device)
        loss, logits, labels = stepper.step_from_raw(raw)
    torch.cuda.synchronize()
    lv = float(loss.detach())
    assert lv == lv and 0 < lv < 100
    assert model.module.queue_ptr.item() == 16


def test_shuffle_overlap_roundtrip_nccl(nccl_world1):
    """begin_shuffle on the comm stream + unshuffle must round-trip."""
    from moco_amd import parallel

    x = torch.randn(16, 128, device="cuda")
    job = parallel.begin_shuffle(x, step=3, seed=7)
    shuffled, idx_unshuffle = job.wait()
    restored = parallel.unshuffle_exchange(shuffled, idx_unshuffle)
    torch.cuda.synchronize()
    assert torch.equal(restored, x)


def test_all_to_all_rows_nccl_world1(nccl_world1):
    from moco_amd import parallel

    x = torch.randn(8, 4, device="cuda")
    perm = torch.randperm(8)
    out = parallel.all_to_all_rows(x, perm)
    torch.cuda.synchronize()
    assert torch.equal(out.cpu(), x.cpu()[torch.argsort(perm)])
