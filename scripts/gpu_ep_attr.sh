#!/usr/bin/env bash
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/ep -- python benchmark/bench_ep_moe.py --iters 30 > /tmp/ep.log 2>&1
grep us/call /tmp/ep.log
python3 scripts/summarize_ktrace.py /tmp/ep gpurun_out/ep_kernels_r02.json
rm -rf /tmp/ep
python3 - <<'PYEOF'
import json
d = json.load(open("gpurun_out/ep_kernels_r02.json"))
for k, v in list(d.items())[:14]:
    print(f"{v['total_ms']:8.2f} ms x{v['calls']:5d}  {k[:60]}")
PYEOF
