#!/usr/bin/env python3
"""GEMM-ReduceScatter overlap benchmark vs the unfused RCCL baseline
(BASELINE config 3). Run: bash scripts/launch.sh 8 benchmark/bench_gemm_rs.py
"""
import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--m", type=int, default=4096)
    p.add_argument("--n", type=int, default=4096)
    p.add_argument("--k", type=int, default=14336)  # sharded FFN down-proj K
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--check", action="store_true")
    args = p.parse_args()

    import triton_dist_amd as td
    from triton_dist_amd.ops import (create_gemm_rs_context, gemm_rs,
                                     gemm_rs_ref)
    from triton_dist_amd.utils import assert_allclose, bf16_gemm_tol, perf_func

    td.initialize_distributed()
    heap = td.init_symm_heap()
    world, rank = heap.world, heap.rank
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    k_shard = args.k // world
    torch.manual_seed(11 + rank)
    a = (torch.randn(args.m, k_shard, device=dev) / 8).to(torch.bfloat16)
    torch.manual_seed(5)
    w = (torch.randn(args.n, k_shard, device=dev) / 8).to(torch.bfloat16)
    ctx = create_gemm_rs_context(args.m, args.n)

    if args.check:
        out = gemm_rs(a, w, ctx)
        ref = gemm_rs_ref(a, w)
        tol = bf16_gemm_tol(k_shard)
        assert_allclose(out, ref, atol=tol["atol"] * world,
                        rtol=tol["rtol"] * 2)
        td.dist_print("check OK")

    out_rs = torch.empty(args.m // world, args.n, dtype=torch.bfloat16,
                         device=dev)

    def torch_path():
        partial = a @ w.t()
        dist.reduce_scatter_tensor(out_rs, partial)
        return out_rs

    def fused_path():
        return gemm_rs(a, w, ctx)

    def gemm_only():
        return a @ w.t()

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
        print(f"GEMM-RS m={args.m} n={args.n} k/shard={k_shard} "
              f"world={world}: torch {r['torch']:.3f} ms | fused "
              f"{r['fused']:.3f} ms (x{speedup:.3f}) | gemm-only "
              f"{r['gemm_only']:.3f} ms")
    td.finalize_distributed()


if __name__ == "__main__":
    main()
