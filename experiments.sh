#!/usr/bin/env bash
# Experiment driver (reference experiments.sh equivalent): runs a batch of
# training sessions, one log per run, clean kill on INT/TERM.
#
# Usage: ./experiments.sh [results-dir]
set -u

RESULTS="${1:-results}"
MAX_STEP="${MAX_STEP:-2000}"   # override for quick smoke runs
mkdir -p "$RESULTS"

PIDS=()
cleanup() {
  for pid in "${PIDS[@]:-}"; do
    kill "$pid" 2>/dev/null || true
  done
  exit 1
}
trap cleanup INT TERM

run() {
  # run <name> <runner args...>
  local name="$1"; shift
  echo "[experiments] $name"
  python3 runner.py "$@" \
    --checkpoint-dir "$RESULTS/$name" \
    --evaluation-delta 50 --evaluation-period -1 \
    --stdout-to "$RESULTS/$name.stdout" --stderr-to "$RESULTS/$name.stderr" &
  local pid=$!
  PIDS+=("$pid")
  wait "$pid" || echo "[experiments] $name FAILED (rc=$?)"
}

GPU_FLAGS=""
if python3 -c 'import torch,sys; sys.exit(0 if torch.cuda.is_available() else 1)'; then
  GPU_FLAGS="--use-gpu --amp"
fi

# Canonical reference config (reference experiments.sh:53): mnist, average,
# n=2, f=0, batch 50. (Step count reduced from the reference's 100k for a
# tractable default; override by editing.)
run mnist-average-n2 \
  --experiment mnist --experiment-args batch-size:50 \
  --aggregator average --nb-workers 2 --max-step $MAX_STEP $GPU_FLAGS

# Robust GARs under data poisoning (mnistAttack, worker 0 poisoned).
run mnistAttack-krum-n5-f1 \
  --experiment mnistAttack --experiment-args batch-size:50 malformed-severity:2 \
  --aggregator krum --nb-workers 5 --nb-decl-byz-workers 1 --max-step $MAX_STEP $GPU_FLAGS

run mnistAttack-average-n5 \
  --experiment mnistAttack --experiment-args batch-size:50 malformed-severity:2 \
  --aggregator average --nb-workers 5 --max-step $MAX_STEP $GPU_FLAGS

# Gradient-reversal attack vs Multi-Krum (BASELINE.json config shape).
run mnist-krum-reversal-n8-f2 \
  --experiment mnist --experiment-args batch-size:50 \
  --aggregator krum --nb-workers 8 --nb-decl-byz-workers 2 \
  --nb-real-byz-workers 2 --attack reversal --max-step $MAX_STEP $GPU_FLAGS

# Lossy-transport (UDP-semantics) + NaN-tolerant GAR.
run mnist-avgnan-lossy-n4 \
  --experiment mnist --experiment-args batch-size:50 \
  --aggregator average-nan --nb-workers 4 \
  --lossy drop-rate:0.05 workers:0 --max-step $MAX_STEP $GPU_FLAGS

echo "[experiments] all done; results in $RESULTS/"
