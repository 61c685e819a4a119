#!/usr/bin/env bash
# Round-3 opening GPU session: the highest-leverage measurements queued
# by round 2 (docs/ROADMAP.md round-3 priorities, docs/PARITY_r02.md
# "remaining gaps"). Usage on a box / via gpurun:
#   bash scripts/r03_first_gpu_session.sh
set -x

# 0. regression anchor: suite + smoke + the three model benches
timeout 600 python -m pytest tests/ -q -m gpu 2>&1 | tail -1
python -c "import __graft_entry__ as g; g.smoke()" | tail -1
timeout 200 python bench.py --steps 12 --warmup 3 | tail -1
timeout 200 python bench.py --model qwen3-30b-a3b --steps 12 --warmup 3 | tail -1

# 1. XCD-granular StaggerU (the r02 per-WG variant broke GROUP_M L2
#    sharing — profiles/README.md; the refinement staggers per L2
#    domain only). A/B once implemented:
TD_GEMM_STAGGER=1 timeout 240 python scripts/bench_stagger.py || true

# 2. GEMM rate attack per the hipBLASLt SK3 ISA study
#    (profiles/hipblaslt_sk3_isa_study.md): prototype the 4-wave
#    128x128-per-wave tile with one direct-to-VGPR operand; numerics
#    gate first, then the ladder:
#    timeout 300 python scripts/bench_stream.py   # shape sweep harness

# 3. If >1 physical GPU is ever available: burn-in FIRST
#    (docs/ROADMAP.md item 1) — the spin overlap paths have never run
#    on real xGMI peers:
# bash scripts/gpu_bench_checks.sh                       # N=2 real GPUs
# TD_AUTOTUNE_METHODS=1 python benchmark/bench_ag_gemm.py  # retune fused
# python benchmark/bench_collectives.py                  # ring-vs-mesh AR

# 4. MoE pq remaining 2x vs weight-stream floor: SRSRC buffer_load
#    staging A/B (ROADMAP item 4) once implemented:
timeout 200 python benchmark/bench_ep_moe.py --iters 40 || true
