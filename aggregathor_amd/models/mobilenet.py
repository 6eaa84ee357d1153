"""MobileNetV2 (capability parity with the reference's slim mobilenet
entries, external/slim/nets/nets_factory.py:39-66)."""

import torch.nn as nn


class InvertedResidual(nn.Module):
    def __init__(self, cin, cout, stride, expand):
        super().__init__()
        hidden = cin * expand
        self.use_res = stride == 1 and cin == cout
        layers = []
        if expand != 1:
            layers += [nn.Conv2d(cin, hidden, 1, bias=False),
                       nn.BatchNorm2d(hidden), nn.ReLU6(inplace=True)]
        layers += [
            nn.Conv2d(hidden, hidden, 3, stride, 1, groups=hidden, bias=False),
            nn.BatchNorm2d(hidden), nn.ReLU6(inplace=True),
            nn.Conv2d(hidden, cout, 1, bias=False), nn.BatchNorm2d(cout)]
        self.conv = nn.Sequential(*layers)

    def forward(self, x):
        out = self.conv(x)
        return x + out if self.use_res else out


class MobileNetV2(nn.Module):
    def __init__(self, num_classes=1000, in_ch=3, width=1.0):
        super().__init__()
        cfg = [  # t, c, n, s
            (1, 16, 1, 1), (6, 24, 2, 2), (6, 32, 3, 2), (6, 64, 4, 2),
            (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]
        c = int(32 * width)
        layers = [nn.Conv2d(in_ch, c, 3, 2, 1, bias=False),
                  nn.BatchNorm2d(c), nn.ReLU6(inplace=True)]
        for t, co, n, s in cfg:
            co = int(co * width)
            for i in range(n):
                layers.append(InvertedResidual(c, co, s if i == 0 else 1, t))
                c = co
        last = int(1280 * max(1.0, width))
        layers += [nn.Conv2d(c, last, 1, bias=False), nn.BatchNorm2d(last),
                   nn.ReLU6(inplace=True)]
        self.features = nn.Sequential(*layers)
        self.pool = nn.AdaptiveAvgPool2d(1)
        self.classifier = nn.Linear(last, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def forward(self, x):
        x = self.pool(self.features(x)).flatten(1)
        return self.classifier(x)


def mobilenet_v2(**kw):
    return MobileNetV2(**kw)
