from .resnet import (
    ResNet,
    BasicBlock,
    Bottleneck,
    build_resnet,
    resnet18,
    resnet34,
    resnet50,
    resnet101,
    resnet152,
)

__all__ = [
    "ResNet",
    "BasicBlock",
    "Bottleneck",
    "build_resnet",
    "resnet18",
    "resnet34",
    "resnet50",
    "resnet101",
    "resnet152",
]
