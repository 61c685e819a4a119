#!/usr/bin/env python3
"""AllReduce latency/bandwidth vs RCCL (one-shot / two-shot heap AR).
Run: bash scripts/launch.sh 8 benchmark/bench_allreduce.py"""
import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--sizes", default="32768,1048576,16777216,134217728")
    p.add_argument("--iters", type=int, default=30)
    args = p.parse_args()

    import triton_dist_amd as td
    from triton_dist_amd.ops import all_reduce, create_allreduce_context
    from triton_dist_amd.utils import perf_func

    td.initialize_distributed()
    heap = td.init_symm_heap()
    world, rank = heap.world, heap.rank
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    sizes = [int(s) for s in args.sizes.split(",")]
    ctx = create_allreduce_context(max(sizes))
    for n in sizes:
        x = torch.randn(n, device=dev).to(torch.bfloat16)
        rows = []
        for name, fn in [
            ("rccl", lambda: dist.all_reduce(x)),
            ("one_shot", lambda: all_reduce(x, ctx, method="one_shot")),
            ("two_shot", lambda: all_reduce(x, ctx, method="two_shot")
             if n % (8 * world) == 0 and world > 1 else None),
        ]:
            if fn is None or (name == "two_shot" and world == 1):
                continue
            try:
                _, ms = perf_func(fn, iters=args.iters, warmup=5)
            except Exception:
                continue
            t = torch.tensor([ms])
            if dist.is_initialized():
                dist.all_reduce(t, op=dist.ReduceOp.MAX)
            gb = n * 2 / 1e9
            rows.append(f"{name} {float(t.item()) * 1e3:9.1f} us "
                        f"({gb / float(t.item()) * 1e3:6.1f} GB/s alg)")
        if rank == 0:
            print(f"AR {n * 2 / 1024 / 1024:8.2f} MiB world={world}: "
                  + " | ".join(rows))
    td.finalize_distributed()


if __name__ == "__main__":
    main()
