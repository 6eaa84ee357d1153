#!/usr/bin/env python3
"""Probe: can concurrent worker streams hide the eager local phase's
launch gaps on the flagship shape?

Since the capture self-check rejects hipGraphs for the imagenet-shape
flagship (profiles/graph_purity_bisect.md), its local phase runs eagerly:
8 sequential micro-batch forward+backwards, ~12 ms/step of launch gaps vs
the captured path. This probe runs the SAME per-worker computation with
k python threads, each submitting its worker's forward+backward on its
own HIP stream via torch.autograd.grad (no shared .grad binding, so
workers are independent; rows are copied into the gradient matrix after).

Prints ms per local phase for k = 1, 2, 4 and the bitwise row agreement
between the k=1 and k>1 runs (the per-worker math is identical and
conv-deterministic WITHIN one process-find; rows must match unless MIOpen
algos are nondeterministic, in which case agreement is reported, not
asserted).
"""

import concurrent.futures
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("PYTORCH_MIOPEN_SUGGEST_NHWC", "1")

import torch

torch.backends.cudnn.benchmark = True


def main():
    from aggregathor_amd import experiments
    from aggregathor_amd.graph import flat_size

    n_workers = 8
    exp = experiments.instantiate(
        "resnet50-imagenet", ["batch-size:32", "eval-examples:0"])
    dev = torch.device("cuda:0")
    torch.manual_seed(1234)
    model = exp.model().to(dev).to(memory_format=torch.channels_last)
    model.train()
    params = [p for p in model.parameters() if p.requires_grad]
    d = flat_size(params)
    rows = torch.zeros((n_workers, d), device=dev)
    batches = [exp.train_batch(w, 0, dev) for w in range(n_workers)]
    batches = [(x.to(memory_format=torch.channels_last), y)
               for x, y in batches]

    def worker_pass(li, stream):
        with torch.cuda.stream(stream):
            with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
                x, y = batches[li]
                loss = exp.loss(model, (x, y))
            grads = torch.autograd.grad(loss, params)
            row = rows[li]
            off = 0
            for g in grads:
                k = g.numel()
                row[off:off + k].copy_(g.reshape(-1))
                off += k

    def phase(k):
        streams = [torch.cuda.Stream() for _ in range(k)]
        main_s = torch.cuda.current_stream()
        for s in streams:
            s.wait_stream(main_s)
        if k == 1:
            for li in range(n_workers):
                worker_pass(li, streams[0])
        else:
            with concurrent.futures.ThreadPoolExecutor(k) as pool:
                futs = [pool.submit(worker_pass, li, streams[li % k])
                        for li in range(n_workers)]
                for f in futs:
                    f.result()
        for s in streams:
            main_s.wait_stream(s)
        torch.cuda.synchronize()

    results = {}
    saved = {}
    for k in (1, 2, 4, 1):
        label = f"k={k}" + ("_repeat" if k == 1 and "k=1" in results else "")
        for _ in range(3):  # warmup (incl. MIOpen find on first)
            phase(k)
        t0 = time.perf_counter()
        iters = 10
        for _ in range(iters):
            phase(k)
        ms = (time.perf_counter() - t0) / iters * 1e3
        results[label] = ms
        saved[label] = rows.clone()
        print(f"local phase {label}: {ms:.2f} ms", flush=True)
    base = saved["k=1"]
    for label, r in saved.items():
        if label == "k=1":
            continue
        same = torch.equal(r, base)
        rel = ((r - base).norm() / base.norm()).item()
        print(f"rows {label} vs k=1: bitwise={same} relL2={rel:.3e}",
              flush=True)


if __name__ == "__main__":
    main()
