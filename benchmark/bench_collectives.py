"""Benchmark the standalone collectives: reduce_scatter, all_to_all_single,
ll_all_gather (latency at small payloads) vs torch.distributed baselines.

Run: bash scripts/launch.sh 8 benchmark/bench_collectives.py [--check]
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist

import triton_dist_amd as td
from triton_dist_amd.ops import (all_to_all_single, create_coll_context,
                                 ll_all_gather, reduce_scatter)


def timeit(fn, iters=50, warmup=5):
    for _ in range(warmup):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    t = torch.tensor([us])
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--check", action="store_true")
    args = ap.parse_args()
    td.initialize_distributed()
    heap = td.init_symm_heap(size_mb=1024)
    world, rank = heap.world, heap.rank
    dev = "cuda" if torch.cuda.is_available() else "cpu"

    ctx = create_coll_context(max_seg_elems=8 << 20, max_ll_words=1 << 14)

    for name, op, shape in [
            ("reduce_scatter", reduce_scatter, (world * 4096, 2048)),
            ("all_to_all", all_to_all_single, (world * 4096, 2048)),
    ]:
        x = torch.randn(shape, device=dev).to(torch.bfloat16)
        us = timeit(lambda: op(x, ctx))
        nbytes = x.numel() * 2
        if rank == 0:
            print(f"{name:16s} {nbytes / 1e6:7.1f} MB  {us:8.1f} us  "
                  f"{nbytes / us / 1e3:6.2f} GB/s (alg)")

    for words in (64, 512, 4096, 16384):
        y = torch.randn(words, device=dev).float().to(torch.bfloat16)
        y = torch.randn(words * 2, device=dev).to(torch.bfloat16)
        us = timeit(lambda: ll_all_gather(y, ctx))
        if rank == 0:
            print(f"ll_all_gather    {words * 4 / 1024:7.1f} KB  "
                  f"{us:8.1f} us")

    if args.check:
        x = torch.randn(world * 256, 512, device=dev).to(torch.bfloat16)
        out = reduce_scatter(x, ctx)
        full = x.float()
        dist.all_reduce(full)
        ref = full.reshape(world, 256, 512)[rank]
        err = (out.float() - ref).abs().max().item()
        print(f"rank {rank}: reduce_scatter check err {err:.3f}")
    td.shutdown_heap()


if __name__ == "__main__":
    main()
