#!/usr/bin/env bash
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/ktr -- python bench.py --steps 4 --warmup 2 > /tmp/attr.log 2>&1
grep tokens_per_s /tmp/attr.log | tail -1
python3 scripts/summarize_ktrace.py /tmp/ktr gpurun_out/dense_kernels_r02_final.json
rm -rf /tmp/ktr
