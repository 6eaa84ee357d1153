"""Normalization-layer selection.

MIOpen's spatial BatchNorm decomposes fwd+bwd into ~6 kernels per layer and
measured ~30% of the ResNet-50 training step (profiles/). PyTorch's native
batch-norm path (cudnn/miopen disabled for the op) uses fewer, fused
kernels. ``AGGREGATHOR_BN=native|miopen`` selects the implementation; the
default is the measured winner on MI355X.
"""

import os

import torch
import torch.nn as nn


class NativeBatchNorm2d(nn.BatchNorm2d):
    """BatchNorm2d forced onto PyTorch's native (non-MIOpen) kernels.

    The per-call ``cudnn.flags`` context costs Python time only outside
    hipGraph capture; inside a captured step it is free at replay.
    """

    def forward(self, x):
        with torch.backends.cudnn.flags(enabled=False):
            return super().forward(x)


def norm2d(channels):
    """BatchNorm2d factory honoring AGGREGATHOR_BN (native | miopen)."""
    kind = os.environ.get("AGGREGATHOR_BN", "miopen")
    if kind == "native":
        return NativeBatchNorm2d(channels)
    return nn.BatchNorm2d(channels)
