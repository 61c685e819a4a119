#!/usr/bin/env bash
# Round-2 opening GPU session: validate everything that landed late in
# r01 compiled-but-unmeasured, in one batch (see docs/ROADMAP.md).
# Usage (on a GPU box / via gpurun): bash scripts/r02_first_gpu_session.sh
set -x

# 1. experimental kernels: numerics first
TD_EXPERIMENTAL=1 timeout 300 python -m pytest tests/test_gpu_kernels.py \
    -q -m gpu -k "gemm256_v2 or paged" -x

# 2. gemm256_v2 A/B vs ring + hipBLASLt (the round-2 headline lever)
timeout 300 python scripts/bench_gemm_v2.py

# 3. megakernel K-split A/B
timeout 300 python benchmark/bench_megakernel.py --model qwen3-8b || true
TD_MK_KSPLIT=4 timeout 300 python benchmark/bench_megakernel.py \
    --model qwen3-8b || true

# 4. hybrid GDN model end-to-end on hardware
timeout 400 python bench.py --model qwen3-next-like --steps 5 --warmup 2

# 5. regression anchor: the r01 headline numbers
timeout 400 python bench.py --steps 10 --warmup 3
timeout 400 python bench.py --model qwen3-30b-a3b --steps 10 --warmup 3
