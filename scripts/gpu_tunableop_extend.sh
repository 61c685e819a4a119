#!/usr/bin/env bash
# Merge seed-oss-36b + qwen3-30b-a3b decode-GEMM tunings into the
# shipped TunableOp file, then measure both models via the shipped
# loader path (no env: tune.maybe_enable_tunableop).
set -x
mkdir -p gpurun_out
cp triton_dist_amd/autotune_cache/tunableop_gfx950.csv gpurun_out/tunableop0.csv
export PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1
export PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tunableop.csv
timeout 240 python bench.py --model seed-oss-36b --ctx 1 --steps 2 --warmup 1 --no-graph 2>&1 | tail -1
timeout 240 python bench.py --model qwen3-30b-a3b --ctx 1 --steps 2 --warmup 1 --no-graph 2>&1 | tail -1
unset PYTORCH_TUNABLEOP_ENABLED PYTORCH_TUNABLEOP_TUNING PYTORCH_TUNABLEOP_FILENAME
# make the merged file the shipped one for THIS box run, then measure
cp gpurun_out/tunableop0.csv triton_dist_amd/autotune_cache/tunableop_gfx950.csv
timeout 200 python bench.py --model seed-oss-36b --steps 6 --warmup 2 2>&1 | tail -1
timeout 200 python bench.py --model qwen3-30b-a3b --steps 10 --warmup 3 2>&1 | tail -1
wc -l gpurun_out/tunableop0.csv
