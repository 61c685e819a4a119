#!/usr/bin/env python3
"""AG-GEMM overlap benchmark vs the unfused RCCL baseline (BASELINE
config 2: Llama-3-8B FFN shapes, M=4096, bf16).

Run: bash scripts/launch.sh 8 benchmark/bench_ag_gemm.py [--check]
Prints per-rank-max latency for:
  torch  : dist.all_gather_into_tensor (RCCL) + torch.matmul (hipBLASLt)
  fused  : ag_gemm (SDMA push producer + per-tile-wait MFMA consumer)
  gemm   : local GEMM only (the compute floor — overlap headroom)
"""
import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--m", type=int, default=4096)       # gathered M
    p.add_argument("--n", type=int, default=14336)      # Llama-3-8B FFN gate
    p.add_argument("--k", type=int, default=4096)
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--check", action="store_true")
    args = p.parse_args()

    import triton_dist_amd as td
    from triton_dist_amd.ops import ag_gemm, ag_gemm_ref, create_ag_gemm_context
    from triton_dist_amd.utils import assert_allclose, bf16_gemm_tol, perf_func

    td.initialize_distributed()
    heap = td.init_symm_heap()
    world, rank = heap.world, heap.rank
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    m_per = args.m // world
    torch.manual_seed(7 + rank)
    a = (torch.randn(m_per, args.k, device=dev) / 8).to(torch.bfloat16)
    torch.manual_seed(99)
    w = (torch.randn(args.n, args.k, device=dev) / 8).to(torch.bfloat16)
    ctx = create_ag_gemm_context(m_per, args.k)

    if args.check:
        c = ag_gemm(a, w, ctx)
        ref = ag_gemm_ref(a, w)
        assert_allclose(c, ref, **bf16_gemm_tol(args.k))
        td.dist_print("check OK")

    full = torch.empty(args.m, args.k, dtype=torch.bfloat16, device=dev)

    def torch_path():
        dist.all_gather_into_tensor(full, a)
        return full @ w.t()

    def fused_path():
        return ag_gemm(a, w, ctx)

    def gemm_only():
        return full @ w.t()

    results = {}
    for name, fn in [("torch", torch_path), ("fused", fused_path),
                     ("gemm_only", gemm_only)]:
        _, ms = perf_func(fn, iters=args.iters, warmup=5)
        t = torch.tensor([ms], device=dev if dev == "cuda" else "cpu")
        if dist.is_initialized():
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
        results[name] = float(t.item())
    if rank == 0:
        r = results
        speedup = r["torch"] / r["fused"] if r["fused"] else 0
        overlap = 0.0
        if r["torch"] > r["gemm_only"]:
            overlap = (r["torch"] - r["fused"]) / (r["torch"] - r["gemm_only"])
        print(f"AG-GEMM m={args.m} n={args.n} k={args.k} world={world}: "
              f"torch {r['torch']:.3f} ms | fused {r['fused']:.3f} ms "
              f"(x{speedup:.3f}) | gemm-only {r['gemm_only']:.3f} ms | "
              f"comm hidden {overlap * 100:.0f}%")
    td.finalize_distributed()


if __name__ == "__main__":
    main()
