#!/usr/bin/env bash
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/ktr -- python bench.py --model qwen3-30b-a3b --steps 4 --warmup 2 > /tmp/attr.log 2>&1
grep tokens_per_s /tmp/attr.log | tail -1
python3 scripts/summarize_ktrace.py /tmp/ktr gpurun_out/moe_kernels_r02.json
rm -rf /tmp/ktr
timeout 300 python bench.py --model qwen3-next-like --steps 6 --warmup 2 2>&1 | tail -1
timeout 420 python bench.py 2>&1 | tail -1
