import torch

from moco_amd.models.resnet import resnet18, resnet50


def test_resnet50_param_count():
    # standard ResNet-50 with a 1000-way fc: 25,557,032 parameters
    m = resnet50(num_classes=1000)
    assert sum(p.numel() for p in m.parameters()) == 25_557_032


def test_resnet18_param_count():
    m = resnet18(num_classes=1000)
    assert sum(p.numel() for p in m.parameters()) == 11_689_512


def test_resnet_forward_shapes():
    m = resnet18(num_classes=128)
    for hw in (32, 64):
        out = m(torch.randn(2, 3, hw, hw))
        assert out.shape == (2, 128)


def test_resnet50_forward_and_grad():
    m = resnet50(num_classes=16)
    out = m(torch.randn(2, 3, 64, 64))
    out.sum().backward()
    assert m.conv1.weight.grad is not None
    assert m.fc.weight.grad is not None


def test_state_dict_key_layout():
    # the checkpoint namespace is a compatibility contract
    keys = set(resnet50(num_classes=10).state_dict().keys())
    for expected in [
        "conv1.weight", "bn1.weight", "bn1.running_mean",
        "layer1.0.conv1.weight", "layer1.0.downsample.0.weight",
        "layer1.0.downsample.1.running_var", "layer4.2.bn3.bias",
        "fc.weight", "fc.bias",
    ]:
        assert expected in keys


def test_grouped_and_wide_variants_param_counts():
    from moco_amd.models.resnet import (resnext50_32x4d, resnext101_32x8d,
                                        wide_resnet50_2)
    # canonical 1000-class parameter counts
    assert sum(p.numel() for p in
               resnext50_32x4d(num_classes=1000).parameters()) == 25_028_904
    assert sum(p.numel() for p in
               wide_resnet50_2(num_classes=1000).parameters()) == 68_883_240
    assert sum(p.numel() for p in
               resnext101_32x8d(num_classes=1000).parameters()) == 88_791_336


def test_resnext_moco_forward():
    import moco_amd
    from moco_amd.models.resnet import resnext50_32x4d
    model = moco_amd.MoCo(
        lambda num_classes: resnext50_32x4d(num_classes=num_classes),
        dim=128, K=64, mlp=True)
    logits, labels = model(torch.randn(4, 3, 64, 64),
                           torch.randn(4, 3, 64, 64))
    assert logits.shape == (4, 65)
