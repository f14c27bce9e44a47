"""ResNet family for the ImageNet workload.

The reference consumed ``torchvision.models.resnet50`` (reference
gossip_sgd.py:693-707); this framework ships its own implementation (the
environment has no torchvision) with identical architecture and the same
initialization recipe the reference applied for large-batch training
("ImageNet in 1 hour"): kaiming-normal convs, zero-init of the final
BatchNorm gamma in every residual branch, and N(0, 0.01) for the FC layer
(reference gossip_sgd.py:698-707).

Layouts: call ``model.to(memory_format=torch.channels_last)`` on MI355X —
MIOpen's NHWC convolutions are the fast path for bf16.
"""

from typing import List, Type, Union

import torch
import torch.nn as nn


# Measured per-shape winners at bs=32/224px on MI355X, fwd+dgrad+wgrad
# summed (profiles/r02_conv_bench2.txt; tools/conv_bench.py): shapes
# where the hand-written MFMA kernels beat MIOpen end to end.  Keys are
# (cin, cout, stride); ResNet-50's channel pairs identify its shapes
# uniquely.  conv_impl='auto' uses MFMA on winners, MIOpen elsewhere.
_MFMA_3X3_WINNERS = {
    (64, 64, 1), (128, 128, 1), (256, 256, 1),
    (512, 512, 2), (512, 512, 1),
}
_MFMA_1X1_WINNERS = {
    (64, 256, 1), (256, 128, 1), (256, 512, 2), (128, 512, 1),
    (512, 128, 1), (512, 256, 1), (512, 1024, 2), (256, 1024, 1),
    (1024, 256, 1), (1024, 512, 1),
}


def conv3x3(cin: int, cout: int, stride: int = 1,
            impl: str = "miopen") -> nn.Conv2d:
    use_mfma = impl == "mfma" or (
        impl == "auto" and (cin, cout, stride) in _MFMA_3X3_WINNERS
    )
    if use_mfma and cin % 64 == 0:
        from .layers import MfmaConv3x3

        return MfmaConv3x3(cin, cout, stride=stride)
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def conv1x1(cin: int, cout: int, stride: int = 1,
            impl: str = "miopen") -> nn.Conv2d:
    if impl == "gemm":
        from .layers import GemmConv1x1

        return GemmConv1x1(cin, cout, stride=stride)
    if impl == "mfma" or (
        impl == "auto" and (cin, cout, stride) in _MFMA_1X1_WINNERS
    ):
        from .layers import MfmaConv1x1

        return MfmaConv1x1(cin, cout, stride=stride)
    return nn.Conv2d(cin, cout, 1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, planes, stride=1, downsample=None,
                 norm_layer=nn.BatchNorm2d, fused=False,
                 conv_impl="miopen"):
        super().__init__()
        self._fused = fused
        self.conv1 = conv3x3(cin, planes, stride, impl=conv_impl)
        self.conv2 = conv3x3(planes, planes, impl=conv_impl)
        self.downsample = downsample
        self.stride = stride
        if fused:
            from .layers import FusedBatchNorm2d

            self.bn1 = FusedBatchNorm2d(planes, relu=True)
            # bn2 fuses the residual add + final relu
            self.bn2 = FusedBatchNorm2d(planes, relu=True)
        else:
            self.bn1 = norm_layer(planes)
            self.bn2 = norm_layer(planes)
            self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        if self._fused:
            out = self.bn1(self.conv1(x))
            identity = x if self.downsample is None else self.downsample(x)
            return self.bn2(self.conv2(out), identity)
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        out += identity
        return self.relu(out)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, planes, stride=1, downsample=None,
                 norm_layer=nn.BatchNorm2d, fused=False,
                 conv_impl="miopen"):
        super().__init__()
        self._fused = fused
        self.conv1 = conv1x1(cin, planes, impl=conv_impl)
        self.conv2 = conv3x3(planes, planes, stride, impl=conv_impl)
        self.conv3 = conv1x1(planes, planes * self.expansion,
                             impl=conv_impl)
        self.downsample = downsample
        self.stride = stride
        if fused:
            from .layers import FusedBatchNorm2d

            self.bn1 = FusedBatchNorm2d(planes, relu=True)
            self.bn2 = FusedBatchNorm2d(planes, relu=True)
            # bn3 fuses the residual add + final relu
            self.bn3 = FusedBatchNorm2d(planes * self.expansion, relu=True)
        else:
            self.bn1 = norm_layer(planes)
            self.bn2 = norm_layer(planes)
            self.bn3 = norm_layer(planes * self.expansion)
            self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        if self._fused:
            out = self.bn1(self.conv1(x))
            out = self.bn2(self.conv2(out))
            identity = x if self.downsample is None else self.downsample(x)
            return self.bn3(self.conv3(out), identity)
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        out += identity
        return self.relu(out)


class ResNet(nn.Module):
    def __init__(
        self,
        block: Type[Union[BasicBlock, Bottleneck]],
        layers: List[int],
        num_classes: int = 1000,
        zero_init_residual: bool = True,
        norm: str = "miopen",
        conv_impl: str = "miopen",
    ):
        super().__init__()
        from .layers import FusedBatchNorm2d, make_norm

        self._fused = norm == "fused"
        self._conv_impl = conv_impl
        self._norm_layer = (
            (lambda c: FusedBatchNorm2d(c, relu=False))
            if self._fused else make_norm(norm)
        )
        self.inplanes = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = (
            FusedBatchNorm2d(64, relu=True) if self._fused
            else self._norm_layer(64)
        )
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * block.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(
                    m.weight, mode="fan_out", nonlinearity="relu"
                )
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

        # large-batch recipe (reference gossip_sgd.py:698-707): zero-init
        # the last BN gamma of each residual branch; fc ~ N(0, 0.01)
        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck):
                    nn.init.zeros_(m.bn3.weight)
                elif isinstance(m, BasicBlock):
                    nn.init.zeros_(m.bn2.weight)
        nn.init.normal_(self.fc.weight, mean=0.0, std=0.01)
        nn.init.zeros_(self.fc.bias)

    def _make_layer(self, block, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride,
                        impl=self._conv_impl),
                self._norm_layer(planes * block.expansion),
            )
        layers = [block(self.inplanes, planes, stride, downsample,
                        norm_layer=self._norm_layer, fused=self._fused,
                        conv_impl=self._conv_impl)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(self.inplanes, planes,
                                norm_layer=self._norm_layer,
                                fused=self._fused,
                                conv_impl=self._conv_impl))
        return nn.Sequential(*layers)

    def forward(self, x):
        if self._fused:
            x = self.maxpool(self.bn1(self.conv1(x)))  # relu fused in bn1
        else:
            x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x)
        x = torch.flatten(x, 1)
        return self.fc(x)


_CONFIGS = {
    "resnet18": (BasicBlock, [2, 2, 2, 2]),
    "resnet34": (BasicBlock, [3, 4, 6, 3]),
    "resnet50": (Bottleneck, [3, 4, 6, 3]),
    "resnet101": (Bottleneck, [3, 4, 23, 3]),
    "resnet152": (Bottleneck, [3, 8, 36, 3]),
}


def build_resnet(
    name: str, num_classes: int = 1000, zero_init_residual: bool = True,
    norm: str = "miopen", conv_impl: str = "miopen",
) -> ResNet:
    block, layers = _CONFIGS[name]
    return ResNet(block, layers, num_classes, zero_init_residual,
                  norm=norm, conv_impl=conv_impl)


def resnet18(**kw) -> ResNet:
    return build_resnet("resnet18", **kw)


def resnet34(**kw) -> ResNet:
    return build_resnet("resnet34", **kw)


def resnet50(**kw) -> ResNet:
    return build_resnet("resnet50", **kw)


def resnet101(**kw) -> ResNet:
    return build_resnet("resnet101", **kw)


def resnet152(**kw) -> ResNet:
    return build_resnet("resnet152", **kw)
