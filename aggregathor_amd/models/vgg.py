"""VGG model family (capability parity with the reference's slim
``vgg_a/16/19`` entries, external/slim/nets/nets_factory.py:39-66).

Standard Simonyan & Zisserman topology with BatchNorm, written directly in
PyTorch.
"""

import torch.nn as nn

_CFGS = {
    "vgg11": [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "vgg16": [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
              512, 512, 512, "M", 512, 512, 512, "M"],
    "vgg19": [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M",
              512, 512, 512, 512, "M", 512, 512, 512, 512, "M"],
}


class VGG(nn.Module):
    def __init__(self, cfg, num_classes=1000, in_ch=3):
        super().__init__()
        layers = []
        c = in_ch
        for v in _CFGS[cfg]:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers += [nn.Conv2d(c, v, 3, padding=1, bias=False),
                           nn.BatchNorm2d(v), nn.ReLU(inplace=True)]
                c = v
        self.features = nn.Sequential(*layers)
        self.avgpool = nn.AdaptiveAvgPool2d(7)
        self.classifier = nn.Sequential(
            nn.Linear(512 * 49, 4096), nn.ReLU(inplace=True), nn.Dropout(),
            nn.Linear(4096, 4096), nn.ReLU(inplace=True), nn.Dropout(),
            nn.Linear(4096, num_classes))
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, 0, 0.01)
                nn.init.zeros_(m.bias)

    def forward(self, x):
        x = self.avgpool(self.features(x)).flatten(1)
        return self.classifier(x)


def vgg11(**kw):
    return VGG("vgg11", **kw)


def vgg16(**kw):
    return VGG("vgg16", **kw)


def vgg19(**kw):
    return VGG("vgg19", **kw)
