"""ResNet backbones, written from scratch for the MI355X-native MoCo engine.

torchvision is not a dependency of this framework: the reference injected
``torchvision.models.resnet50`` as the encoder (reference ``main_moco.py:165``);
here we carry our own implementation.  Parameter/buffer names intentionally
follow the de-facto ResNet state-dict layout (``conv1``, ``bn1``,
``layerX.Y.convZ`` ...) because the checkpoint namespace is a compatibility
contract (reference ``main_moco.py:348-352``, ``main_lincls.py:172-190`` and
``detection/convert-pretrain-to-detectron2.py`` all parse it).

Only the architecture *shape* is standard; the module code is ours and is
organized so individual blocks can be swapped for fused HIP implementations
(conv+BN+ReLU block kernels) without changing the state-dict layout.
"""

from typing import Callable, List, Optional, Type, Union

import torch
import torch.nn as nn

from ..ops.block import fused_bottleneck
from ..ops.bn import bn_act
from ..ops.conv1x1 import conv1x1 as conv1x1_op
from ..ops.pool import head_avgpool, stem_maxpool


def conv3x3(cin: int, cout: int, stride: int = 1, dilation: int = 1,
            groups: int = 1) -> nn.Conv2d:
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=dilation,
                     dilation=dilation, groups=groups, bias=False)


def conv1x1(cin: int, cout: int, stride: int = 1) -> nn.Conv2d:
    return nn.Conv2d(cin, cout, 1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes: int, planes: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None,
                 norm_layer: Callable[..., nn.Module] = nn.BatchNorm2d):
        super().__init__()
        self.conv1 = conv3x3(inplanes, planes, stride)
        self.bn1 = norm_layer(planes)
        self.relu = nn.ReLU(inplace=True)
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = norm_layer(planes)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        identity = x
        out = bn_act(self.conv1(x), self.bn1, relu=True)
        if self.downsample is not None:
            identity = bn_act(self.downsample[0](x), self.downsample[1],
                              relu=False)
        return bn_act(self.conv2(out), self.bn2, relu=True,
                      residual=identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes: int, planes: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None,
                 norm_layer: Callable[..., nn.Module] = nn.BatchNorm2d,
                 groups: int = 1, base_width: int = 64):
        super().__init__()
        width = int(planes * (base_width / 64.0)) * groups
        self.conv1 = conv1x1(inplanes, width)
        self.bn1 = norm_layer(width)
        self.conv2 = conv3x3(width, width, stride, groups=groups)
        self.bn2 = norm_layer(width)
        self.conv3 = conv1x1(width, planes * self.expansion)
        self.bn3 = norm_layer(planes * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # hot path: the whole block as fused MFMA GEMMs with BN stats/apply
        # folded into prologues/epilogues (ops/block.py)
        out = fused_bottleneck(x, self)
        if out is not None:
            return out
        identity = x
        out = bn_act(conv1x1_op(x, self.conv1), self.bn1, relu=True)
        out = bn_act(self.conv2(out), self.bn2, relu=True)
        if self.downsample is not None:
            identity = bn_act(self.downsample[0](x), self.downsample[1],
                              relu=False)
        return bn_act(conv1x1_op(out, self.conv3), self.bn3, relu=True,
                      residual=identity)


class ResNet(nn.Module):
    def __init__(self, block: Type[Union[BasicBlock, Bottleneck]],
                 layers: List[int], num_classes: int = 1000,
                 zero_init_residual: bool = False,
                 norm_layer: Callable[..., nn.Module] = nn.BatchNorm2d,
                 groups: int = 1, width_per_group: int = 64):
        super().__init__()
        self._norm_layer = norm_layer
        self.groups = groups
        self.base_width = width_per_group
        self.inplanes = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = norm_layer(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(kernel_size=3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(512 * block.expansion, num_classes)

        # wire predecessor references across the trunk: a fused Bottleneck
        # can take over its predecessor's bn3 backward-reduce (ops/block.py
        # cross-block dgrad-epilogue fusion)
        if block is Bottleneck:
            trunk = [b for layer in (self.layer1, self.layer2, self.layer3,
                                     self.layer4) for b in layer]
            for prev_b, b in zip(trunk[:-1], trunk[1:]):
                # bypass nn.Module.__setattr__: a plain reference, NOT a
                # registered child module (state_dict must not see it)
                object.__setattr__(b, "_prev_bottleneck", prev_b)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, (nn.BatchNorm2d, nn.GroupNorm)):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck):
                    nn.init.constant_(m.bn3.weight, 0)
                elif isinstance(m, BasicBlock):
                    nn.init.constant_(m.bn2.weight, 0)

    def _make_layer(self, block, planes: int, blocks: int,
                    stride: int = 1) -> nn.Sequential:
        norm_layer = self._norm_layer
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                norm_layer(planes * block.expansion),
            )
        kw = {}
        if block is Bottleneck:
            kw = dict(groups=self.groups, base_width=self.base_width)
        layers = [block(self.inplanes, planes, stride, downsample,
                        norm_layer, **kw)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(self.inplanes, planes,
                                norm_layer=norm_layer, **kw))
        return nn.Sequential(*layers)

    def _update_bn_counters(self):
        """One foreach add for every BN's num_batches_tracked instead of
        ~53 single-element kernels per forward (bn_act/fused_bottleneck
        skip their own increments for counters tagged here).  Rebuilt
        lazily because .to()/.cuda() replace buffer tensors."""
        cs = getattr(self, "_bn_counters", None)
        if cs is None or cs[0] is not self.bn1.num_batches_tracked:
            cs = []
            for m in self.modules():
                if isinstance(m, nn.modules.batchnorm._BatchNorm) \
                        and m.num_batches_tracked is not None:
                    cs.append(m.num_batches_tracked)
                    m._nbt_managed = True
            self._bn_counters = cs
        if cs:
            torch._foreach_add_(cs, 1)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.training:
            self._update_bn_counters()
        x = bn_act(self.conv1(x), self.bn1, relu=True)
        x = stem_maxpool(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = head_avgpool(x)
        x = torch.flatten(x, 1)
        return self.fc(x)


def resnet18(**kwargs) -> ResNet:
    return ResNet(BasicBlock, [2, 2, 2, 2], **kwargs)


def resnet34(**kwargs) -> ResNet:
    return ResNet(BasicBlock, [3, 4, 6, 3], **kwargs)


def resnet50(**kwargs) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], **kwargs)


def resnet101(**kwargs) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 23, 3], **kwargs)


def resnet152(**kwargs) -> ResNet:
    return ResNet(Bottleneck, [3, 8, 36, 3], **kwargs)


def resnext50_32x4d(**kwargs) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], groups=32, width_per_group=4,
                  **kwargs)


def resnext101_32x8d(**kwargs) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 23, 3], groups=32, width_per_group=8,
                  **kwargs)


def wide_resnet50_2(**kwargs) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], width_per_group=128, **kwargs)


MODEL_REGISTRY = {
    "resnet18": resnet18,
    "resnet34": resnet34,
    "resnet50": resnet50,
    "resnet101": resnet101,
    "resnet152": resnet152,
    "resnext50_32x4d": resnext50_32x4d,
    "resnext101_32x8d": resnext101_32x8d,
    "wide_resnet50_2": wide_resnet50_2,
}


def build_model(arch: str, **kwargs) -> ResNet:
    if arch not in MODEL_REGISTRY:
        raise ValueError(
            f"unknown arch {arch!r}; available: {sorted(MODEL_REGISTRY)}")
    return MODEL_REGISTRY[arch](**kwargs)
