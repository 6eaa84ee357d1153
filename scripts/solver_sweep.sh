#!/bin/bash
# Identify the capture-unsafe MIOpen solver family: run the smallest
# drifting purity cases under per-family disables + solver logging.
CASES="conv1x1_64-64_8px"
run() {
  echo "##### $1"
  shift
  env "$@" timeout 300 python scripts/graph_purity_bisect.py \
      --filter "$CASES" 2>&1 | grep -E "PURE|DRIFT|ERROR|config"
}
# Name the chosen solvers for the failing conv (fwd/bwd-data/bwd-weights).
echo "##### solver-log conv1x1_64-64_8px"
MIOPEN_ENABLE_LOGGING=1 MIOPEN_LOG_LEVEL=5 timeout 300 \
  python scripts/graph_purity_bisect.py --filter conv1x1_64-64_8px 2>&1 \
  | grep -iE "solution|solver|algorithm|PURE|DRIFT" | sort | uniq -c | sort -rn | head -40

run baseline
run no-benchmark AGGREGATHOR_BISECT_BENCHMARK=0
run deterministic AGGREGATHOR_DETERMINISTIC_CONV=1
run no-gemm MIOPEN_DEBUG_CONV_GEMM=0
run no-direct MIOPEN_DEBUG_CONV_DIRECT=0
run no-winograd MIOPEN_DEBUG_CONV_WINOGRAD=0
run no-implicit-gemm MIOPEN_DEBUG_CONV_IMPLICIT_GEMM=0
run no-ck MIOPEN_DEBUG_CONV_CK_IGEMM_FWD_V6R1_DLOPS_NCHW=0
run no-wino-no-igemm MIOPEN_DEBUG_CONV_WINOGRAD=0 MIOPEN_DEBUG_CONV_IMPLICIT_GEMM=0
