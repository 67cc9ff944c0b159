from .resnet import (MODEL_REGISTRY, ResNet, build_model, resnet18, resnet34,
                     resnet50, resnet101, resnet152, resnext50_32x4d,
                     resnext101_32x8d, wide_resnet50_2)

__all__ = [
    "ResNet",
    "MODEL_REGISTRY",
    "build_model",
    "resnet18",
    "resnet34",
    "resnet50",
    "resnet101",
    "resnet152",
    "resnext50_32x4d",
    "resnext101_32x8d",
    "wide_resnet50_2",
]
