#!/usr/bin/env bash
# Single-node multi-GPU launcher (capability parity with the reference's
# scripts/launch.sh torchrun wrapper — Triton-distributed
# scripts/launch.sh:160-180, launch_amd.sh).
#
# Usage: scripts/launch.sh <nproc> <script.py> [args...]
set -euo pipefail

NPROC=${1:?usage: launch.sh <nproc> <script.py> [args...]}
shift

export MASTER_ADDR=${MASTER_ADDR:-127.0.0.1}
export MASTER_PORT=${MASTER_PORT:-$((20000 + RANDOM % 20000))}
# dmabuf IPC is required for hipIpc across processes on this driver stack
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}
# keep kernel args in device memory (fewer launch stalls)
export HIP_FORCE_DEV_KERNARG=${HIP_FORCE_DEV_KERNARG:-1}

# Sanitizer/debugger hook (reference parity: launch.sh:162-164 TORCHRUN
# override): TD_LAUNCH_WRAPPER="rocgdb --batch -ex run --args" or
# a sanitizer runner wraps every rank's interpreter.
exec ${TD_LAUNCH_WRAPPER:-} python -m torch.distributed.run \
  --nnodes=1 --nproc-per-node "$NPROC" \
  --master-addr "$MASTER_ADDR" --master-port "$MASTER_PORT" \
  "$@"
