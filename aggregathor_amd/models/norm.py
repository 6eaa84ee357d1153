"""Normalization-layer selection.

MIOpen's spatial BatchNorm decomposes fwd+bwd into ~6 kernels per layer and
measured ~30% of the ResNet-50 training step (profiles/). PyTorch's native
batch-norm path (cudnn/miopen disabled for the op) uses fewer, fused
kernels. ``AGGREGATHOR_BN=native|miopen`` selects the implementation; the
default is the measured winner on MI355X.

STATUS of the from-scratch kernels (AGGREGATHOR_BN=fused,
ops/csrc/bn_kernels.hip): NUMERICS TESTBED, not a hot-path component.
After three measured optimization rounds (NOTES.md) they reached MIOpen
PARITY (0.112 vs 0.103 ms/layer-pass; 69.1 vs 67.5 ms full step) but not
a win, so the default stays ``miopen`` and the fused path is kept for
what it uniquely provides: a fully-controlled, deterministic,
fp64-referenced BatchNorm implementation used by the numerics tests
(tests/test_gpu_bn.py) -- including the shifted-variance formulation that
documents (and guards against) the catastrophic E[x^2]-E[x]^2
cancellation. It is not selected by any default configuration.
"""

import os

import torch
import torch.nn as nn


class _FusedBNFunction(torch.autograd.Function):
    """Training-mode batch norm on the fused gfx950 kernels (ops/csrc/
    bn_kernels.hip): 2+2 main memory passes instead of MIOpen's 6 kernels.
    Running-stat updates happen inside the forward kernel; gradients for
    weight/bias are fp32 (accumulate into the flattened gradient views)."""

    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, momentum,
                eps):
        from ..ops import _load_extension
        ext = _load_extension()
        y, mean, invstd = ext.bn_fwd_train(
            x, weight.float(), bias.float(), running_mean, running_var,
            momentum, eps)
        ctx.save_for_backward(x, weight, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        from ..ops import _load_extension
        ext = _load_extension()
        x, weight, mean, invstd = ctx.saved_tensors
        dx, dw, db = ext.bn_bwd_train(dy, x, weight.float(), mean, invstd)
        return (dx, dw.to(weight.dtype), db.to(weight.dtype),
                None, None, None, None)


class FusedBatchNorm2d(nn.BatchNorm2d):
    """BatchNorm2d using the fused NHWC HIP kernels when applicable
    (training, GPU, channels_last, C % 4 == 0, bf16/fp32); falls back to
    the stock implementation otherwise (eval mode, CPU, odd shapes).

    EXPERIMENTAL (AGGREGATHOR_BN=fused, default off): variance is computed
    single-pass SHIFTED by the running mean (exact where mean drift
    develops; the naive E[x^2]-E[x]^2 form diverged on ResNet-50 @ 224).
    MIOpen's BN measured faster per layer (profiles/NOTES.md), so this path
    is a numerics-tested starting point for fusion work, not the
    default."""

    def forward(self, x):
        if (self.training and x.is_cuda and x.dim() == 4
                and x.size(1) % 4 == 0
                and x.dtype in (torch.float32, torch.bfloat16)
                and self.affine and self.track_running_stats
                and self.momentum is not None):
            x = x.contiguous(memory_format=torch.channels_last)
            return _FusedBNFunction.apply(
                x, self.weight, self.bias, self.running_mean,
                self.running_var, self.momentum, self.eps)
        return super().forward(x)


class NativeBatchNorm2d(nn.BatchNorm2d):
    """BatchNorm2d forced onto PyTorch's native (non-MIOpen) kernels.

    The per-call ``cudnn.flags`` context costs Python time only outside
    hipGraph capture; inside a captured step it is free at replay.
    """

    def forward(self, x):
        with torch.backends.cudnn.flags(enabled=False):
            return super().forward(x)


def norm2d(channels):
    """BatchNorm2d factory honoring AGGREGATHOR_BN (fused | native | miopen)."""
    kind = os.environ.get("AGGREGATHOR_BN", "miopen")
    if kind == "fused":  # EXPERIMENTAL -- see FusedBatchNorm2d docstring
        return FusedBatchNorm2d(channels)
    if kind == "native":
        return NativeBatchNorm2d(channels)
    return nn.BatchNorm2d(channels)
