#!/usr/bin/env bash
# PMC counter evidence (own run: --pmc must not mix with trace domains).
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
timeout 420 rocprofv3 --pmc SQ_INSTS_MFMA SQ_BUSY_CYCLES SQ_WAIT_ANY SQ_LDS_BANK_CONFLICT -d gpurun_out/pmc_moe -- python bench.py --model qwen3-30b-a3b --steps 3 --warmup 1 > gpurun_out/pmc_moe.log 2>&1
tail -2 gpurun_out/pmc_moe.log
ls gpurun_out/pmc_moe/* 2>/dev/null | head -3
