#!/usr/bin/env bash
# PMC counter evidence: counters-only run, summarized ON the box so only
# a small JSON travels back (raw per-dispatch dbs exceed the pull cap).
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
timeout 420 rocprofv3 --pmc SQ_INSTS_MFMA SQ_BUSY_CYCLES SQ_WAIT_ANY SQ_LDS_BANK_CONFLICT -d /tmp/pmcout -- python bench.py --model qwen3-30b-a3b --steps 2 --warmup 1 > /tmp/pmc.log 2>&1
tail -1 /tmp/pmc.log
python3 scripts/summarize_pmc.py /tmp/pmcout gpurun_out/pmc_moe_r02.json
rm -rf /tmp/pmcout
tail -c 1200 gpurun_out/pmc_moe_r02.json
