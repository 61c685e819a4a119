#!/usr/bin/env bash
# Produce a real on-hardware autotune cache (ag method + gemm_ar sk at
# the 2-rank test shapes) and pull it back for shipping in the repo.
set -x
mkdir -p gpurun_out/autotune
export TD_AUTOTUNE_DIR=$GRAFT_REPO_ROOT/gpurun_out/autotune
export TD_AUTOTUNE_METHODS=1
timeout 400 python -m pytest tests/test_gpu_dist.py::test_ag_gemm_2rank tests/test_gpu_allreduce.py::test_gemm_ar_tiled_gpu_2rank -q -m gpu 2>&1 | tail -2
find gpurun_out/autotune -name "*.json" | head; cat gpurun_out/autotune/*/*.json 2>/dev/null | head -40
