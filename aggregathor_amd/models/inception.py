"""Inception model family, written from the paper topologies.

Capability parity with the reference's slim ``networks_map`` entries
``inception_v1`` and ``inception_v3``
(/root/reference/external/slim/nets/nets_factory.py:39-66):

- Inception v1 (GoogLeNet): Szegedy et al., "Going Deeper with
  Convolutions" (2014), with batch norm after every convolution (as slim's
  inception_v1 does).
- Inception v3: Szegedy et al., "Rethinking the Inception Architecture for
  Computer Vision" (2015): factorized 7x7 (17x17 grid), asymmetric
  convolutions, efficient grid reductions, expanded 8x8 blocks.

Both are input-size tolerant (global average pool head), so they register
against the same imagenet/cifar10 dataset shapes as the other families.
The auxiliary classifier heads are omitted (documented scope cut,
PARITY.md): the reference's aux-logits loss applies a 0.4/0.3-weighted
side loss during training only; the primary topology and parameter shapes
are complete without them.
"""

import torch
import torch.nn as nn


class _ConvBN(nn.Module):
    """conv + BN + ReLU, the slim conv2d default for inception nets."""

    def __init__(self, cin, cout, kernel, stride=1, padding=0):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, kernel, stride=stride,
                              padding=padding, bias=False)
        self.bn = nn.BatchNorm2d(cout, eps=1e-3)
        self.act = nn.ReLU(inplace=True)

    def forward(self, x):
        return self.act(self.bn(self.conv(x)))


# --------------------------------------------------------------------------- #
# Inception v1 (GoogLeNet)


class _InceptionV1Block(nn.Module):
    """The classic 4-branch block: 1x1 / 1x1-3x3 / 1x1-5x5 / pool-1x1."""

    def __init__(self, cin, c1, c3r, c3, c5r, c5, pp):
        super().__init__()
        self.b1 = _ConvBN(cin, c1, 1)
        self.b2 = nn.Sequential(_ConvBN(cin, c3r, 1),
                                _ConvBN(c3r, c3, 3, padding=1))
        self.b3 = nn.Sequential(_ConvBN(cin, c5r, 1),
                                _ConvBN(c5r, c5, 5, padding=2))
        self.b4 = nn.Sequential(nn.MaxPool2d(3, stride=1, padding=1),
                                _ConvBN(cin, pp, 1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], 1)


class InceptionV1(nn.Module):
    """GoogLeNet: stem + 9 inception blocks (3a..5b) + GAP head."""

    def __init__(self, num_classes=1000, in_ch=3):
        super().__init__()
        self.stem = nn.Sequential(
            _ConvBN(in_ch, 64, 7, stride=2, padding=3),
            nn.MaxPool2d(3, stride=2, ceil_mode=True),
            _ConvBN(64, 64, 1),
            _ConvBN(64, 192, 3, padding=1),
            nn.MaxPool2d(3, stride=2, ceil_mode=True),
        )
        self.i3a = _InceptionV1Block(192, 64, 96, 128, 16, 32, 32)
        self.i3b = _InceptionV1Block(256, 128, 128, 192, 32, 96, 64)
        self.pool3 = nn.MaxPool2d(3, stride=2, ceil_mode=True)
        self.i4a = _InceptionV1Block(480, 192, 96, 208, 16, 48, 64)
        self.i4b = _InceptionV1Block(512, 160, 112, 224, 24, 64, 64)
        self.i4c = _InceptionV1Block(512, 128, 128, 256, 24, 64, 64)
        self.i4d = _InceptionV1Block(512, 112, 144, 288, 32, 64, 64)
        self.i4e = _InceptionV1Block(528, 256, 160, 320, 32, 128, 128)
        self.pool4 = nn.MaxPool2d(3, stride=2, ceil_mode=True)
        self.i5a = _InceptionV1Block(832, 256, 160, 320, 32, 128, 128)
        self.i5b = _InceptionV1Block(832, 384, 192, 384, 48, 128, 128)
        self.head = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(1024, num_classes)

    def forward(self, x):
        x = self.stem(x)
        x = self.pool3(self.i3b(self.i3a(x)))
        x = self.pool4(self.i4e(self.i4d(self.i4c(self.i4b(self.i4a(x))))))
        x = self.i5b(self.i5a(x))
        x = self.head(x).flatten(1)
        return self.fc(x)


# --------------------------------------------------------------------------- #
# Inception v3


class _InceptionA(nn.Module):
    """35x35 block: 1x1 / 5x5 / double-3x3 / pool branches."""

    def __init__(self, cin, pool_features):
        super().__init__()
        self.b1 = _ConvBN(cin, 64, 1)
        self.b5 = nn.Sequential(_ConvBN(cin, 48, 1),
                                _ConvBN(48, 64, 5, padding=2))
        self.b3d = nn.Sequential(_ConvBN(cin, 64, 1),
                                 _ConvBN(64, 96, 3, padding=1),
                                 _ConvBN(96, 96, 3, padding=1))
        self.bp = nn.Sequential(nn.AvgPool2d(3, stride=1, padding=1),
                                _ConvBN(cin, pool_features, 1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b5(x), self.b3d(x), self.bp(x)], 1)


class _ReductionA(nn.Module):
    """35x35 -> 17x17 grid reduction."""

    def __init__(self, cin):
        super().__init__()
        self.b3 = _ConvBN(cin, 384, 3, stride=2)
        self.b3d = nn.Sequential(_ConvBN(cin, 64, 1),
                                 _ConvBN(64, 96, 3, padding=1),
                                 _ConvBN(96, 96, 3, stride=2))
        self.pool = nn.MaxPool2d(3, stride=2)

    def forward(self, x):
        return torch.cat([self.b3(x), self.b3d(x), self.pool(x)], 1)


class _InceptionC(nn.Module):
    """17x17 block with factorized 7x7 (1x7 + 7x1) convolutions."""

    def __init__(self, cin, c7):
        super().__init__()
        self.b1 = _ConvBN(cin, 192, 1)
        self.b7 = nn.Sequential(
            _ConvBN(cin, c7, 1),
            _ConvBN(c7, c7, (1, 7), padding=(0, 3)),
            _ConvBN(c7, 192, (7, 1), padding=(3, 0)))
        self.b7d = nn.Sequential(
            _ConvBN(cin, c7, 1),
            _ConvBN(c7, c7, (7, 1), padding=(3, 0)),
            _ConvBN(c7, c7, (1, 7), padding=(0, 3)),
            _ConvBN(c7, c7, (7, 1), padding=(3, 0)),
            _ConvBN(c7, 192, (1, 7), padding=(0, 3)))
        self.bp = nn.Sequential(nn.AvgPool2d(3, stride=1, padding=1),
                                _ConvBN(cin, 192, 1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b7(x), self.b7d(x), self.bp(x)], 1)


class _ReductionB(nn.Module):
    """17x17 -> 8x8 grid reduction."""

    def __init__(self, cin):
        super().__init__()
        self.b3 = nn.Sequential(_ConvBN(cin, 192, 1),
                                _ConvBN(192, 320, 3, stride=2))
        self.b7x3 = nn.Sequential(
            _ConvBN(cin, 192, 1),
            _ConvBN(192, 192, (1, 7), padding=(0, 3)),
            _ConvBN(192, 192, (7, 1), padding=(3, 0)),
            _ConvBN(192, 192, 3, stride=2))
        self.pool = nn.MaxPool2d(3, stride=2)

    def forward(self, x):
        return torch.cat([self.b3(x), self.b7x3(x), self.pool(x)], 1)


class _InceptionE(nn.Module):
    """8x8 block with expanded 1x3/3x1 filter banks."""

    def __init__(self, cin):
        super().__init__()
        self.b1 = _ConvBN(cin, 320, 1)
        self.b3_stem = _ConvBN(cin, 384, 1)
        self.b3_a = _ConvBN(384, 384, (1, 3), padding=(0, 1))
        self.b3_b = _ConvBN(384, 384, (3, 1), padding=(1, 0))
        self.b3d_stem = nn.Sequential(_ConvBN(cin, 448, 1),
                                      _ConvBN(448, 384, 3, padding=1))
        self.b3d_a = _ConvBN(384, 384, (1, 3), padding=(0, 1))
        self.b3d_b = _ConvBN(384, 384, (3, 1), padding=(1, 0))
        self.bp = nn.Sequential(nn.AvgPool2d(3, stride=1, padding=1),
                                _ConvBN(cin, 192, 1))

    def forward(self, x):
        s = self.b3_stem(x)
        d = self.b3d_stem(x)
        return torch.cat([
            self.b1(x),
            torch.cat([self.b3_a(s), self.b3_b(s)], 1),
            torch.cat([self.b3d_a(d), self.b3d_b(d)], 1),
            self.bp(x)], 1)


class InceptionV3(nn.Module):
    """Inception v3 (299x299 canonical; any input >= 75px works, smaller
    inputs degrade to 1x1 grids through the ceil-less pools)."""

    def __init__(self, num_classes=1000, in_ch=3, dropout=0.5):
        super().__init__()
        self.stem = nn.Sequential(
            _ConvBN(in_ch, 32, 3, stride=2),
            _ConvBN(32, 32, 3),
            _ConvBN(32, 64, 3, padding=1),
            nn.MaxPool2d(3, stride=2, ceil_mode=True),
            _ConvBN(64, 80, 1),
            _ConvBN(80, 192, 3),
            nn.MaxPool2d(3, stride=2, ceil_mode=True),
        )
        self.mixed = nn.Sequential(
            _InceptionA(192, 32),
            _InceptionA(256, 64),
            _InceptionA(288, 64),
            _ReductionA(288),
            _InceptionC(768, 128),
            _InceptionC(768, 160),
            _InceptionC(768, 160),
            _InceptionC(768, 192),
            _ReductionB(768),
            _InceptionE(1280),
            _InceptionE(2048),
        )
        self.head = nn.AdaptiveAvgPool2d(1)
        self.dropout = nn.Dropout(dropout)
        self.fc = nn.Linear(2048, num_classes)

    def forward(self, x):
        x = self.stem(x)
        x = self.mixed(x)
        x = self.head(x).flatten(1)
        return self.fc(self.dropout(x))


def inception_v1(num_classes=1000, **kw):
    return InceptionV1(num_classes=num_classes, **kw)


def inception_v3(num_classes=1000, **kw):
    return InceptionV3(num_classes=num_classes, **kw)
