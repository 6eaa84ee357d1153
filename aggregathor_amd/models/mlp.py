"""784-100-10 MLP (reference experiments/mnist.py:83-104,132)."""

import torch.nn as nn


class MLP(nn.Module):
    """Fully-connected ReLU network; default dims match the reference MNIST
    experiment's ``_inference([784, 100, 10], ...)``."""

    def __init__(self, dims=(784, 100, 10)):
        super().__init__()
        layers = []
        for i in range(len(dims) - 1):
            layers.append(nn.Linear(dims[i], dims[i + 1]))
            if i < len(dims) - 2:
                layers.append(nn.ReLU(inplace=True))
        self.net = nn.Sequential(*layers)

    def forward(self, x):
        if x.dim() > 2:
            x = x.flatten(1)
        return self.net(x)
