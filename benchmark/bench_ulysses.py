#!/usr/bin/env python3
"""Fused Ulysses pair (qkv-GEMM<->a2a, a2a<->o-GEMM) vs the unfused
GEMM + standalone a2a composition.
Run: bash scripts/launch.sh 8 benchmark/bench_ulysses.py [--check]"""
import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--tokens", type=int, default=2048)
    p.add_argument("--hidden", type=int, default=1024)
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--check", action="store_true")
    args = p.parse_args()

    import triton_dist_amd as td
    from triton_dist_amd.ops import (create_ulysses_fused_context,
                                     ulysses_a2a_o_gemm,
                                     ulysses_qkv_gemm_a2a)
    from triton_dist_amd.utils import assert_allclose, dist_print, perf_func

    td.initialize_distributed()
    heap = td.init_symm_heap()
    world, rank = heap.world, heap.rank
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    t_loc, hdim = args.tokens, args.hidden
    qkv_dim = world * 1280
    o_in = 1024
    n_out = hdim
    ctx = create_ulysses_fused_context(t_loc, qkv_dim, o_in)
    torch.manual_seed(3 + rank)
    x = (torch.randn(t_loc, hdim, device=dev) / 8).to(torch.bfloat16)
    torch.manual_seed(42)
    w_qkv = (torch.randn(qkv_dim, hdim, device=dev) / 8).to(torch.bfloat16)
    w_o_split = (torch.randn(world, n_out, o_in, device=dev) / 8
                 ).to(torch.bfloat16)
    if args.check:
        mine = ulysses_qkv_gemm_a2a(x, w_qkv, ctx)
        xs = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(xs, x)
        full = torch.cat(xs, 0).float() @ w_qkv.float().t()
        pc = qkv_dim // world
        assert_allclose(mine, full[:, rank * pc:(rank + 1) * pc]
                        .to(torch.bfloat16), atol=2.5e-1, rtol=5e-2)
        dist_print("[check ok] qkv_gemm_a2a")
        return
    ms = perf_func(lambda: ulysses_qkv_gemm_a2a(x, w_qkv, ctx),
                   iters=args.iters, warmup=5)
    dist_print(f"qkv_gemm_a2a T_loc={t_loc}: {ms*1e3:8.1f} us")
    attn = ulysses_qkv_gemm_a2a(x, w_qkv, ctx)[:, :o_in].contiguous()
    ms = perf_func(lambda: ulysses_a2a_o_gemm(attn, w_o_split, ctx),
                   iters=args.iters, warmup=5)
    dist_print(f"a2a_o_gemm   T_loc={t_loc}: {ms*1e3:8.1f} us")


if __name__ == "__main__":
    main()
