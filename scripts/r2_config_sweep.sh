#!/bin/bash
# Re-measure the BASELINE.json config table on the VERIFIED execution
# path (round 2: capture self-check + hybrid/eager fallback).
set -u
mkdir -p gpurun_out
J() { grep -h "steps/sec" "$1" | tail -1; }

# B: MNIST MLP, Krum f=2, n=8 (runner, GPU)
timeout 300 python runner.py --experiment mnist --experiment-args batch-size:50 \
  --aggregator krum --nb-workers 8 --nb-decl-byz-workers 2 --max-step 2000 \
  --use-gpu --evaluation-delta -1 --evaluation-period -1 --summary-delta -1 \
  --summary-period -1 --checkpoint-delta -1 --checkpoint-period -1 \
  --evaluation-file - > gpurun_out/cfgB.log 2>&1; echo B=$?
grep "Steps/s" gpurun_out/cfgB.log

# C: CIFAR-10 ResNet-20, Bulyan f=1, n=8, batch 64
timeout 300 python bench.py --model resnet20 --dataset cifar10 --gar bulyan \
  --f 1 --batch-size 64 --steps 100 --warmup 5 > gpurun_out/cfgC.log 2>&1; echo C=$?
J gpurun_out/cfgC.log

# D: ResNet-50 Multi-Krum f=2 + reversal attack
timeout 400 python bench.py --attack reversal --steps 20 --warmup 5 \
  > gpurun_out/cfgD.log 2>&1; echo D=$?
J gpurun_out/cfgD.log

# E: ResNet-200, Bulyan f=2 n=11, lossy drop
timeout 500 python bench.py --model resnet200 --workers 11 --gar bulyan \
  --steps 10 --warmup 3 --lossy drop-rate:0.01 workers:0 \
  > gpurun_out/cfgE.log 2>&1; echo E=$?
J gpurun_out/cfgE.log

# Model sweep (valid path)
for M in resnet18 resnet101 vgg16 mobilenet_v2; do
  timeout 400 python bench.py --model $M --steps 10 --warmup 4 \
    > gpurun_out/sweep_$M.log 2>&1; echo $M=$?
  J gpurun_out/sweep_$M.log
done

# Runner soak: 1200 captured resnet20 steps WITH services + checkpoint,
# then resume 300 more (checkpoint/resume on GPU).
CKPT=/tmp/soak_ckpt
timeout 500 python runner.py --experiment resnet20-cifar10 \
  --experiment-args batch-size:32 --aggregator krum --nb-workers 8 \
  --nb-decl-byz-workers 2 --max-step 1200 --use-gpu --amp \
  --checkpoint-dir $CKPT --checkpoint-delta 500 --checkpoint-period -1 \
  --evaluation-delta 400 --evaluation-period -1 --summary-delta 200 \
  --summary-period -1 > gpurun_out/soak1.log 2>&1; echo soak1=$?
timeout 300 python runner.py --experiment resnet20-cifar10 \
  --experiment-args batch-size:32 --aggregator krum --nb-workers 8 \
  --nb-decl-byz-workers 2 --max-step 300 --use-gpu --amp \
  --checkpoint-dir $CKPT --checkpoint-delta 500 --checkpoint-period -1 \
  --evaluation-delta 400 --evaluation-period -1 --summary-delta 200 \
  --summary-period -1 > gpurun_out/soak2.log 2>&1; echo soak2=$?
grep -E "Steps/s|Restored|Saved|diverged|top1" gpurun_out/soak1.log | tail -5
grep -E "Steps/s|Restored|Saved|diverged|top1" gpurun_out/soak2.log | tail -5
