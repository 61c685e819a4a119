#!/usr/bin/env bash
# Round-2 GPU session 1: validate the late-r01 experimental work.
#   1. gemm256_v2 numerics + A/B vs ring + hipBLASLt  (VERDICT next-1)
#   2. paged flash-decode numerics
#   3. megakernel K-split A/B                         (VERDICT next-5)
#   4. bench anchor (qwen3-32b decode, regression check)
set -x
mkdir -p gpurun_out
TD_EXPERIMENTAL=1 timeout 240 python -m pytest tests/test_gpu_kernels.py \
    -q -m gpu -k "gemm256_v2 or paged" -x 2>&1 | tail -20
timeout 300 python scripts/bench_gemm_v2.py 2>&1 | tee gpurun_out/gemm_v2_ab.log | tail -10
timeout 240 python benchmark/bench_megakernel.py --model qwen3-8b 2>&1 | tee gpurun_out/mk_base.log | tail -15
TD_MK_KSPLIT=4 timeout 240 python benchmark/bench_megakernel.py --model qwen3-8b 2>&1 | tee gpurun_out/mk_ks4.log | tail -15
TD_MK_KSPLIT=8 timeout 240 python benchmark/bench_megakernel.py --model qwen3-8b 2>&1 | tee gpurun_out/mk_ks8.log | tail -15
timeout 300 python bench.py --steps 10 --warmup 3 2>&1 | tee gpurun_out/bench_anchor.log | tail -3
