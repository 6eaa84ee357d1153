"""CIFAR-10 CNN (reference experiments/cnnet.py:58-95).

Architecture parity: conv5x5(3->64, SAME) + ReLU + maxpool3x3/2(SAME),
conv5x5(64->64, SAME) + ReLU + maxpool3x3/2(SAME), dense 384 + ReLU,
dense 192 + ReLU, linear 10. Initializers mirror the reference's
truncated-normal/constant choices.
"""

import torch
import torch.nn as nn


def _trunc_normal_(t, std):
    nn.init.trunc_normal_(t, std=std, a=-2 * std, b=2 * std)


class CNNet(nn.Module):
    def __init__(self, num_classes=10, in_ch=3, image_size=32):
        super().__init__()
        self.conv1 = nn.Conv2d(in_ch, 64, 5, padding=2)
        self.conv2 = nn.Conv2d(64, 64, 5, padding=2)
        # TF 'SAME' 3x3/2 pooling on even sizes keeps ceil(n/2): pad right/bottom.
        self.pool = nn.MaxPool2d(3, stride=2, padding=1)
        feat = (image_size // 4) ** 2 * 64
        self.fc3 = nn.Linear(feat, 384)
        self.fc4 = nn.Linear(384, 192)
        self.fc5 = nn.Linear(192, num_classes)
        with torch.no_grad():
            _trunc_normal_(self.conv1.weight, 5e-2)
            nn.init.zeros_(self.conv1.bias)
            _trunc_normal_(self.conv2.weight, 5e-2)
            nn.init.constant_(self.conv2.bias, 0.1)
            _trunc_normal_(self.fc3.weight, 0.04)
            nn.init.constant_(self.fc3.bias, 0.1)
            _trunc_normal_(self.fc4.weight, 0.04)
            nn.init.constant_(self.fc4.bias, 0.1)
            _trunc_normal_(self.fc5.weight, 1 / 192.0)
            nn.init.zeros_(self.fc5.bias)
        self.act = nn.ReLU(inplace=True)

    def forward(self, x):
        x = self.pool(self.act(self.conv1(x)))
        x = self.pool(self.act(self.conv2(x)))
        x = x.flatten(1)
        x = self.act(self.fc3(x))
        x = self.act(self.fc4(x))
        return self.fc5(x)
