#!/usr/bin/env bash
# Round-2 evidence batch: FA2 A/B, rocprof decode attribution, full suite.
set -x
mkdir -p gpurun_out
timeout 300 python scripts/bench_prefill_attn.py 2>&1 | tee gpurun_out/prefill_attn_r02.log | tail -6
cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r02 -- python bench.py --steps 5 --warmup 2 > gpurun_out/bench_prof_r02.log 2>&1
tail -3 gpurun_out/bench_prof_r02.log
ls gpurun_out/prof_r02 2>/dev/null | head -5
timeout 900 python -m pytest tests/ -q -m gpu 2>&1 | tail -4
