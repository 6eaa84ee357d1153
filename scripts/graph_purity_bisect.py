#!/usr/bin/env python3
"""Localize the capture-unsafe kernel behind the hipGraph divergence.

The engine-level purity check (graph_divergence_repro.py --case purity)
proved the captured resnet50-cifar10 local phase is NOT replay-pure:
replaying the captured graph twice on frozen inputs drifts the produced
gradients by ~1e33 (gpurun_out/purity.log). This harness captures
fwd+bwd graphs of individual submodules / single ops on the exact shapes
the flagship configs use and replays each twice: the smallest case that
drifts names the culprit kernel, and the dims that matter (dtype, memory
format, grad-view binding, spatial size) name the gate.

Usage: python scripts/graph_purity_bisect.py [--filter substr]
Prints one PURE/DRIFT line per case.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("PYTORCH_MIOPEN_SUGGEST_NHWC", "1")

import torch
import torch.nn as nn

torch.backends.cudnn.benchmark = \
    os.environ.get("AGGREGATHOR_BISECT_BENCHMARK", "1") == "1"
if os.environ.get("AGGREGATHOR_DETERMINISTIC_CONV") == "1":
    torch.backends.cudnn.deterministic = True


def purity_case(name, make_module, input_shape, amp=True,
                channels_last=True, grad_views=True, replays=3):
    """Capture fwd+bwd of module(input) and replay `replays` times on
    identical state; returns max bitwise drift between replays."""
    import contextlib
    from aggregathor_amd.graph import bind_grad_views, flat_size
    torch.manual_seed(1234)
    dev = torch.device("cuda:0")
    model = make_module().to(dev)
    if channels_last and any(p.dim() == 4 for p in model.parameters()):
        model = model.to(memory_format=torch.channels_last)
    model.train()
    params = [p for p in model.parameters() if p.requires_grad]
    d = flat_size(params)
    row = torch.zeros(d, device=dev)
    x = torch.randn(input_shape, device=dev)
    if channels_last and x.dim() == 4:
        x = x.to(memory_format=torch.channels_last)

    def one_pass():
        amp_ctx = (torch.autocast(device_type="cuda", dtype=torch.bfloat16)
                   if amp else contextlib.nullcontext())
        row.zero_()
        if grad_views:
            bind_grad_views(params, row)
        else:
            for p in params:
                p.grad = torch.zeros_like(p)
        with amp_ctx:
            out = model(x)
            loss = out.float().square().mean()
        loss.backward()
        return loss.detach()

    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):  # warmup (allocator + MIOpen find)
        for _ in range(2):
            one_pass()
    torch.cuda.current_stream().wait_stream(side)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        one_pass()

    def snap():
        if grad_views:
            return row.clone()
        return torch.cat([p.grad.reshape(-1) for p in params])

    g.replay()
    torch.cuda.synchronize()
    ref = snap()
    worst = 0.0
    for _ in range(replays - 1):
        g.replay()
        torch.cuda.synchronize()
        cur = snap()
        finite = torch.isfinite(cur).all().item()
        drift = (cur - ref).abs().max().item() if finite else float("inf")
        worst = max(worst, drift)
    status = "PURE " if worst == 0.0 else "DRIFT"
    print(f"{status} {name:42s} d={d:>9d} max_drift={worst:.3e}", flush=True)
    del g
    return worst


def conv_bn_cases():
    """Shapes from resnet50-cifar10 (the diverging config, batch 32,
    input 3x32x32): stem 7x7/2 -> 16x16, maxpool -> 8x8, stages at
    8/4/2/1 spatial."""
    cases = []

    def conv(cin, cout, k, s=1):
        return lambda: nn.Conv2d(cin, cout, k, stride=s, padding=k // 2,
                                 bias=False)

    def convbnrelu(cin, cout, k, s=1):
        def make():
            return nn.Sequential(
                nn.Conv2d(cin, cout, k, stride=s, padding=k // 2, bias=False),
                nn.BatchNorm2d(cout), nn.ReLU(inplace=True))
        return make

    B = 32
    cases += [
        ("stem_conv7x7s2_3-64_32px", conv(3, 64, 7, 2), (B, 3, 32, 32)),
        ("conv1x1_64-64_8px", conv(64, 64, 1), (B, 64, 8, 8)),
        ("conv3x3_64-64_8px", conv(64, 64, 3), (B, 64, 8, 8)),
        ("conv1x1_64-256_8px", conv(64, 256, 1), (B, 64, 8, 8)),
        ("conv1x1_256-512_8px_s2", conv(256, 512, 1, 2), (B, 256, 8, 8)),
        ("conv3x3_128-128_4px", conv(128, 128, 3), (B, 128, 4, 4)),
        ("conv3x3_256-256_2px", conv(256, 256, 3), (B, 256, 2, 2)),
        ("conv1x1_1024-2048_2px_s2", conv(1024, 2048, 1, 2), (B, 1024, 2, 2)),
        ("conv3x3_512-512_1px", conv(512, 512, 3), (B, 512, 1, 1)),
        ("conv1x1_2048-512_1px", conv(2048, 512, 1), (B, 2048, 1, 1)),
        ("bn64_8px", lambda: nn.BatchNorm2d(64), (B, 64, 8, 8)),
        ("bn2048_1px", lambda: nn.BatchNorm2d(2048), (B, 2048, 1, 1)),
        ("cbr3x3_64-64_8px", convbnrelu(64, 64, 3), (B, 64, 8, 8)),
        ("cbr1x1_2048-512_1px", convbnrelu(2048, 512, 1), (B, 2048, 1, 1)),
        ("linear2048-1000", lambda: nn.Linear(2048, 1000), (B, 2048)),
    ]
    return cases


def model_cases():
    from aggregathor_amd.models import NETWORKS
    B = 32

    def slice_of(arch, attr):
        def make():
            m = NETWORKS[arch](num_classes=10)
            return getattr(m, attr)
        return make

    def whole(arch, classes=10):
        return lambda: NETWORKS[arch](num_classes=classes)

    return [
        ("r50_layer1_8px", slice_of("resnet50", "layer1"), (B, 64, 8, 8)),
        ("r50_layer2_8px", slice_of("resnet50", "layer2"), (B, 256, 8, 8)),
        ("r50_layer3_4px", slice_of("resnet50", "layer3"), (B, 512, 4, 4)),
        ("r50_layer4_2px", slice_of("resnet50", "layer4"), (B, 1024, 2, 2)),
        ("resnet50_full_cifar", whole("resnet50"), (B, 3, 32, 32)),
        ("resnet20_full_cifar", whole("resnet20"), (B, 3, 32, 32)),
    ]


def imagenet_cases():
    """The flagship bench shapes (resnet50-imagenet, batch 32/worker):
    stage inputs at 56/28/14/7 px -- the shapes the headline number rides."""
    from aggregathor_amd.models import NETWORKS
    B = 32

    def slice_of(arch, attr):
        def make():
            m = NETWORKS[arch](num_classes=1000)
            return getattr(m, attr)
        return make

    return [
        ("IM_r50_layer1_56px", slice_of("resnet50", "layer1"), (B, 64, 56, 56)),
        ("IM_r50_layer2_28px", slice_of("resnet50", "layer2"), (B, 256, 56, 56)),
        ("IM_r50_layer3_14px", slice_of("resnet50", "layer3"), (B, 512, 28, 28)),
        ("IM_r50_layer4_7px", slice_of("resnet50", "layer4"), (B, 1024, 14, 14)),
        ("IM_resnet50_full", lambda: NETWORKS["resnet50"](num_classes=1000),
         (B, 3, 224, 224)),
    ]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--filter", default="")
    ap.add_argument("--amp", default="1")
    ap.add_argument("--views", default="1")
    ap.add_argument("--channels-last", dest="cl", default="1")
    ap.add_argument("--replays", type=int, default=3)
    ap.add_argument("--set", dest="case_set", default="cifar",
                    choices=["cifar", "imagenet", "all"])
    args = ap.parse_args()
    amp = args.amp == "1"
    views = args.views == "1"
    cl = args.cl == "1"
    print(f"config: amp={amp} grad_views={views} channels_last={cl} "
          f"set={args.case_set}", flush=True)
    cases = []
    if args.case_set in ("cifar", "all"):
        cases += conv_bn_cases() + model_cases()
    if args.case_set in ("imagenet", "all"):
        cases += imagenet_cases()
    drifted = []
    for name, make, shape in cases:
        if args.filter and args.filter not in name:
            continue
        try:
            w = purity_case(name, make, shape, amp=amp, channels_last=cl,
                            grad_views=views, replays=args.replays)
            if w != 0.0:
                drifted.append(name)
        except RuntimeError as e:
            print(f"ERROR {name}: {str(e)[:200]}", flush=True)
            drifted.append(name + " (error)")
    print("\n=== drifting cases ===")
    for n in drifted:
        print(" ", n)
    if not drifted:
        print("  none")


if __name__ == "__main__":
    main()
