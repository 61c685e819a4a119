#!/usr/bin/env python3
"""Tile-granular fused GEMM+AllReduce vs sequential GEMM + AR vs
GEMM + RCCL all_reduce (the gemm_ar TP mode's core op).
Run: bash scripts/launch.sh 8 benchmark/bench_gemm_ar.py [--check]"""
import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--shapes",
                   default="512x5120x3456,2048x5120x1024,2048x5120x640")
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--check", action="store_true")
    args = p.parse_args()

    import triton_dist_amd as td
    from triton_dist_amd.ops import (all_reduce, create_allreduce_context,
                                     gemm_allreduce)
    from triton_dist_amd.ops.allreduce import _gemm_ar_tiled_hip
    from triton_dist_amd.ops.gemm import gemm
    from triton_dist_amd.utils import assert_allclose, dist_print, perf_func

    td.initialize_distributed()
    heap = td.init_symm_heap()
    world, rank = heap.world, heap.rank
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    shapes = [tuple(int(v) for v in s.split("x"))
              for s in args.shapes.split(",")]
    ctx = create_allreduce_context(max(m * n for m, n, _ in shapes))
    for m, n, k in shapes:
        torch.manual_seed(11 + rank)
        a = (torch.randn(m, k, device=dev) / 8).to(torch.bfloat16)
        w = (torch.randn(n, k, device=dev) / 8).to(torch.bfloat16)
        if args.check:
            c = gemm_allreduce(a, w, ctx)
            ref = a.float() @ w.float().t()
            dist.all_reduce(ref)
            assert_allclose(c, ref.to(torch.bfloat16), atol=2.5e-1,
                            rtol=5e-2)
            dist_print(f"[check ok] {m}x{n}x{k}")
            continue
        rows = []
        runs = [("fused", lambda: gemm_allreduce(a, w, ctx))]
        if heap.backend == "hip" and world > 1:
            runs.append(("seq gemm+AR",
                         lambda: all_reduce(gemm(a, w), ctx)))

            def _rccl():
                c = gemm(a, w)
                dist.all_reduce(c)
                return c

            runs.append(("gemm+rccl", _rccl))
        for name, fn in runs:
            ms = perf_func(fn, iters=args.iters, warmup=5)
            rows.append(f"{name} {ms*1e3:8.1f} us")
        dist_print(f"{m}x{n}x{k}: " + " | ".join(rows))


if __name__ == "__main__":
    main()
