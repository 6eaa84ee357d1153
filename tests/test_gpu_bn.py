"""Fused NHWC BatchNorm kernels vs fp64 autograd reference on MI355X."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from aggregathor_amd import ops
from aggregathor_amd.models.norm import FusedBatchNorm2d


def _require_ext():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    ext = ops._load_extension()
    assert ext is not None and hasattr(ext, "bn_fwd_train")
    return ext


def _reference(x64, w64, b64, eps):
    """fp64 autograd reference: y, dx, dw, db, batch mean/var."""
    x64 = x64.detach().requires_grad_(True)
    w64 = w64.detach().requires_grad_(True)
    b64 = b64.detach().requires_grad_(True)
    mean = x64.mean(dim=(0, 2, 3))
    var = x64.var(dim=(0, 2, 3), unbiased=False)
    xhat = (x64 - mean[None, :, None, None]) / torch.sqrt(
        var[None, :, None, None] + eps)
    y = xhat * w64[None, :, None, None] + b64[None, :, None, None]
    return x64, w64, b64, y, mean, var


@pytest.mark.parametrize("shape,dtype", [
    ((8, 64, 56, 56), torch.bfloat16),
    ((8, 64, 56, 56), torch.float32),
    ((4, 2048, 7, 7), torch.bfloat16),   # wide-channel regime (C >= 1024)
    ((16, 24, 14, 14), torch.bfloat16),  # C not a power of two
    ((2, 16, 5, 3), torch.float32),      # odd spatial
])
def test_bn_fwd_bwd_vs_reference(shape, dtype):
    ext = _require_ext()
    torch.manual_seed(shape[1])
    eps, momentum = 1e-5, 0.1
    n, c, h, w = shape
    x = (torch.randn(shape, device="cuda") * 2 + 0.5).to(dtype) \
        .contiguous(memory_format=torch.channels_last)
    weight = torch.rand(c, device="cuda") + 0.5
    bias = torch.randn(c, device="cuda")
    running_mean = torch.zeros(c, device="cuda")
    running_var = torch.ones(c, device="cuda")

    y, mean, invstd = ext.bn_fwd_train(
        x, weight, bias, running_mean.clone(), running_var.clone(),
        momentum, eps)

    x64, w64, b64, y_ref, mean_ref, var_ref = _reference(
        x.double(), weight.double(), bias.double(), eps)

    is_bf16 = dtype == torch.bfloat16
    tol = dict(rtol=2e-2, atol=2e-2) if is_bf16 else dict(rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(mean.double().cpu(), mean_ref.cpu(),
                               rtol=1e-3, atol=1e-3)
    torch.testing.assert_close(y.float().cpu(), y_ref.float().cpu(), **tol)

    # Backward vs fp64 autograd.
    dy = torch.randn(shape, device="cuda").to(dtype) \
        .contiguous(memory_format=torch.channels_last)
    dx, dw, db = ext.bn_bwd_train(dy, x, weight, mean, invstd)
    y_ref.backward(dy.double())
    torch.testing.assert_close(dx.float().cpu(), x64.grad.float().cpu(), **tol)
    torch.testing.assert_close(dw.double().cpu(), w64.grad.cpu(),
                               rtol=1e-2 if is_bf16 else 1e-3, atol=1e-1)
    torch.testing.assert_close(db.double().cpu(), b64.grad.cpu(),
                               rtol=1e-2 if is_bf16 else 1e-3, atol=1e-1)


def test_bn_running_stats_update():
    ext = _require_ext()
    torch.manual_seed(3)
    c, momentum, eps = 32, 0.1, 1e-5
    x = torch.randn(4, c, 8, 8, device="cuda") \
        .contiguous(memory_format=torch.channels_last)
    rm = torch.full((c,), 0.5, device="cuda")
    rv = torch.full((c,), 2.0, device="cuda")
    ref = torch.nn.BatchNorm2d(c, momentum=momentum, eps=eps).cuda()
    with torch.no_grad():
        ref.running_mean.fill_(0.5)
        ref.running_var.fill_(2.0)
    ext.bn_fwd_train(x, ref.weight.detach().clone(), ref.bias.detach().clone(),
                     rm, rv, momentum, eps)
    ref(x)  # torch path updates its own running stats
    torch.testing.assert_close(rm, ref.running_mean, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(rv, ref.running_var, rtol=1e-4, atol=1e-5)


def test_fused_module_trains():
    _require_ext()
    torch.manual_seed(5)
    bn = FusedBatchNorm2d(32).cuda()
    opt = torch.optim.SGD(bn.parameters(), lr=0.1)
    for _ in range(4):
        x = torch.randn(8, 32, 16, 16, device="cuda",
                        requires_grad=True).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        with torch.autocast("cuda", torch.bfloat16):
            y = bn(x)
        loss = y.float().square().mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
    assert torch.isfinite(bn.weight).all()
    # Eval mode falls back to the stock implementation.
    bn.eval()
    out = bn(torch.randn(2, 32, 16, 16, device="cuda"))
    assert torch.isfinite(out).all()
