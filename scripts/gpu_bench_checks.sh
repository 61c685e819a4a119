#!/usr/bin/env bash
# benchmark/*.py --check sweep at N=2 (one GPU, hipIpc sharing) — the
# N=2 slice of the VERDICT multi-GPU burn-in item.
set -x
RUN="python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1"
port=29611
for b in bench_ag_gemm bench_gemm_rs bench_gemm_ar bench_collectives \
         bench_ulysses bench_ep_moe bench_gdn bench_pp bench_tp_layers; do
  port=$((port+1))
  timeout 300 $RUN --master-port $port benchmark/$b.py --check 2>&1 \
    | grep -E "check|ok|PASS|Error|Trace" | tail -3
done
