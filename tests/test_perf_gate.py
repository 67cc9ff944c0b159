"""Opt-in performance regression gate.

Run on an MI355X with MOCO_AMD_PERF_GATE=1: a short flagship-config
measurement must clear a generous floor (large regressions — naive-conv
fallbacks, broken fusion dispatch — trip it; box-to-box variance does not).
Round-1 measured 5,843 img/s at this config; the gate is 3,000.
"""

import os
import time

import pytest
import torch

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(os.environ.get("MOCO_AMD_PERF_GATE") != "1",
                       reason="opt-in: MOCO_AMD_PERF_GATE=1"),
]


def test_flagship_step_throughput_floor():
    import moco_amd
    from moco_amd.engine import MoCoStepper
    from moco_amd.models.resnet import resnet50
    from moco_amd.ops import FusedSGD

    device = torch.device("cuda", 0)
    torch.manual_seed(0)
    model = moco_amd.MoCo(
        lambda num_classes: resnet50(num_classes=num_classes),
        dim=128, K=65536, m=0.999, T=0.2, mlp=True)
    model = model.to(device).to(memory_format=torch.channels_last)
    opt = FusedSGD(model.parameters(), lr=0.03, momentum=0.9,
                   weight_decay=1e-4)
    stepper = MoCoStepper(model, opt, device, aug_plus=True, out_size=224)
    raw = torch.rand(256, 3, 256, 256, device=device)
    for _ in range(6):  # warmup absorbs the MIOpen find
        stepper.step_from_raw(raw)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5):
        stepper.step_from_raw(raw)
    torch.cuda.synchronize()
    img_s = 256 * 5 / (time.perf_counter() - t0)
    assert img_s > 3000, f"throughput regression: {img_s:.0f} img/s"
