#!/usr/bin/env python3
"""Decisive probe: is an EAGER ALLOCATION between hipGraph replays the
trigger for the capture corruption?

Everything observed so far fits this hypothesis (see
profiles/graph_purity_bisect.md): every checking protocol that detected
garbage allocated device memory between replays (clone of the 800 MB row
matrix), while the healthy production loops (bench hot loop, the r1
2000-step resnet20 run) allocate nothing between replays; the r1 runner
divergences all followed allocating events (service threads, the repro's
step-100 max|p| print).

Protocols per case (buffers preallocated before the first replay):
  noalloc      replay -> copy_ -> replay -> copy_      (only eager kernels)
  clone        replay -> clone(row) -> replay           (alloc + free)
  alloc_free   replay -> empty(256MB); del -> replay    (alloc + free)
  alloc_hold   replay -> empty(256MB) kept  -> replay   (alloc, no free)
Reports the replay-to-replay drift under each.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("PYTORCH_MIOPEN_SUGGEST_NHWC", "1")

import torch
import torch.nn as nn

torch.backends.cudnn.benchmark = True


def build_graph(make_module, input_shape, amp=True):
    import contextlib
    from aggregathor_amd.graph import bind_grad_views, flat_size
    torch.manual_seed(1234)
    dev = torch.device("cuda:0")
    model = make_module().to(dev)
    if any(p.dim() == 4 for p in model.parameters()):
        model = model.to(memory_format=torch.channels_last)
    model.train()
    params = [p for p in model.parameters() if p.requires_grad]
    row = torch.zeros(flat_size(params), device=dev)
    x = torch.randn(input_shape, device=dev)
    if x.dim() == 4:
        x = x.to(memory_format=torch.channels_last)

    def one_pass():
        ctx = (torch.autocast(device_type="cuda", dtype=torch.bfloat16)
               if amp else contextlib.nullcontext())
        row.zero_()
        bind_grad_views(params, row)
        with ctx:
            loss = model(x).float().square().mean()
        loss.backward()

    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for _ in range(2):
            one_pass()
    torch.cuda.current_stream().wait_stream(side)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        one_pass()
    return g, row


def run_case(name, make_module, input_shape):
    # Preallocate the comparison buffers BEFORE capture: an allocation
    # anywhere after capture is itself a candidate trigger.
    import contextlib
    from aggregathor_amd.graph import flat_size
    probe_model = make_module()
    d = flat_size([p for p in probe_model.parameters() if p.requires_grad])
    del probe_model
    ref1 = torch.empty(d, device="cuda:0")
    ref2 = torch.empty(d, device="cuda:0")
    g, row = build_graph(make_module, input_shape)

    def drift(protocol):
        g.replay()
        torch.cuda.synchronize()
        ref1.copy_(row)
        held = None
        if protocol == "clone":
            _ = row.clone()
        elif protocol == "alloc_free":
            junk = torch.empty(64 * 1024 * 1024, device=row.device)
            del junk
        elif protocol == "alloc_hold":
            held = torch.empty(64 * 1024 * 1024, device=row.device)
        torch.cuda.synchronize()
        g.replay()
        torch.cuda.synchronize()
        ref2.copy_(row)
        del held
        finite = bool(torch.isfinite(ref2).all())
        d = (ref2 - ref1).abs().max().item() if finite else float("inf")
        return d

    out = {}
    for protocol in ("noalloc", "clone", "alloc_free", "alloc_hold",
                     "noalloc"):
        key = protocol if protocol not in out else protocol + "_2"
        out[key] = drift(protocol)
    print(f"[{name}] " + "  ".join(f"{k}={v:.3e}" for k, v in out.items()),
          flush=True)
    del g
    return out


def main():
    run_case("conv1x1_64-64_8px",
             lambda: nn.Conv2d(64, 64, 1, bias=False), (32, 64, 8, 8))
    from aggregathor_amd.models import NETWORKS
    run_case("resnet50_imagenet_b8",
             lambda: NETWORKS["resnet50"](num_classes=1000),
             (8, 3, 224, 224))


if __name__ == "__main__":
    main()
