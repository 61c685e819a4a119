#!/usr/bin/env bash
# benchmark/*.py --check sweep at N=2 (one GPU, hipIpc sharing) — the
# N=2 slice of the VERDICT multi-GPU burn-in item. Full per-bench logs
# land in gpurun_out/checks/<bench>.log so failures carry tracebacks.
RUN="python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1"
mkdir -p gpurun_out/checks
port=29611
BENCHES="${BENCHES:-bench_ag_gemm bench_gemm_rs bench_gemm_ar bench_collectives bench_ulysses bench_ep_moe bench_gdn bench_pp bench_tp_layers}"
for b in $BENCHES; do
  port=$((port+1))
  timeout 280 $RUN --master-port $port benchmark/$b.py --check \
    > gpurun_out/checks/$b.log 2>&1
  rc=$?
  echo "== $b rc=$rc"
  grep -E "check|OK|Error" gpurun_out/checks/$b.log | grep -v Warning | tail -2
done
