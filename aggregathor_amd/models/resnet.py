"""ResNet v1 model family.

Capability parity with the reference's vendored TF-slim nets
(external/slim/nets/resnet_v1.py:281+, nets_factory.py:39-66): ImageNet-shape
resnet18/34/50/101/152/200 plus the CIFAR-style resnet20/32/44/56/110.
Written from the standard ResNet paper topology (He et al. 2015) in idiomatic
PyTorch -- NOT a port of the slim code.

All models use channels-last-friendly conv stacks; on MI355X they are run
under bf16 autocast with fp32 master weights and fp32 flattened gradients
(the GAR input dtype matches the reference's float kernels).
"""

import torch.nn as nn

from .norm import norm2d


def _conv3(cin, cout, stride=1):
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def _conv1(cin, cout, stride=1):
    return nn.Conv2d(cin, cout, 1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = _conv3(cin, planes, stride)
        self.bn1 = norm2d(planes)
        self.conv2 = _conv3(planes, planes)
        self.bn2 = norm2d(planes)
        self.act = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        idn = x if self.downsample is None else self.downsample(x)
        out = self.act(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        return self.act(out + idn)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = _conv1(cin, planes)
        self.bn1 = norm2d(planes)
        self.conv2 = _conv3(planes, planes, stride)
        self.bn2 = norm2d(planes)
        self.conv3 = _conv1(planes, planes * 4)
        self.bn3 = norm2d(planes * 4)
        self.act = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        idn = x if self.downsample is None else self.downsample(x)
        out = self.act(self.bn1(self.conv1(x)))
        out = self.act(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        return self.act(out + idn)


class ResNet(nn.Module):
    """ImageNet-shape ResNet v1 (7x7/2 stem + 3x3/2 maxpool + 4 stages)."""

    def __init__(self, block, layers, num_classes=1000, in_ch=3):
        super().__init__()
        self.inplanes = 64
        self.conv1 = nn.Conv2d(in_ch, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = norm2d(64)
        self.act = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._stage(block, 64, layers[0])
        self.layer2 = self._stage(block, 128, layers[1], stride=2)
        self.layer3 = self._stage(block, 256, layers[2], stride=2)
        self.layer4 = self._stage(block, 512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * block.expansion, num_classes)
        self._init_weights()

    def _stage(self, block, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                _conv1(self.inplanes, planes * block.expansion, stride),
                norm2d(planes * block.expansion))
        stage = [block(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            stage.append(block(self.inplanes, planes))
        return nn.Sequential(*stage)

    def _init_weights(self):
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def forward(self, x):
        x = self.maxpool(self.act(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


class CifarResNet(nn.Module):
    """CIFAR-style ResNet (3x3 stem, 3 stages of 16/32/64, 6n+2 layers)."""

    def __init__(self, depth, num_classes=10, in_ch=3):
        super().__init__()
        assert (depth - 2) % 6 == 0, "CIFAR ResNet depth must be 6n+2"
        n = (depth - 2) // 6
        self.inplanes = 16
        self.conv1 = _conv3(in_ch, 16)
        self.bn1 = norm2d(16)
        self.act = nn.ReLU(inplace=True)
        self.layer1 = self._stage(16, n)
        self.layer2 = self._stage(32, n, stride=2)
        self.layer3 = self._stage(64, n, stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(64, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _stage(self, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes:
            downsample = nn.Sequential(
                _conv1(self.inplanes, planes, stride), norm2d(planes))
        stage = [BasicBlock(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes
        for _ in range(1, blocks):
            stage.append(BasicBlock(self.inplanes, planes))
        return nn.Sequential(*stage)

    def forward(self, x):
        x = self.act(self.bn1(self.conv1(x)))
        x = self.layer3(self.layer2(self.layer1(x)))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet18(**kw):
    return ResNet(BasicBlock, [2, 2, 2, 2], **kw)


def resnet34(**kw):
    return ResNet(BasicBlock, [3, 4, 6, 3], **kw)


def resnet50(**kw):
    return ResNet(Bottleneck, [3, 4, 6, 3], **kw)


def resnet101(**kw):
    return ResNet(Bottleneck, [3, 4, 23, 3], **kw)


def resnet152(**kw):
    return ResNet(Bottleneck, [3, 8, 36, 3], **kw)


def resnet200(**kw):
    return ResNet(Bottleneck, [3, 24, 36, 3], **kw)


def cifar_resnet20(**kw):
    return CifarResNet(20, **kw)


def cifar_resnet32(**kw):
    return CifarResNet(32, **kw)


def cifar_resnet44(**kw):
    return CifarResNet(44, **kw)


def cifar_resnet56(**kw):
    return CifarResNet(56, **kw)


def cifar_resnet110(**kw):
    return CifarResNet(110, **kw)


RESNETS = {
    "resnet18": resnet18, "resnet34": resnet34, "resnet50": resnet50,
    "resnet101": resnet101, "resnet152": resnet152, "resnet200": resnet200,
    "resnet20": cifar_resnet20, "resnet32": cifar_resnet32,
    "resnet44": cifar_resnet44, "resnet56": cifar_resnet56,
    "resnet110": cifar_resnet110,
}
