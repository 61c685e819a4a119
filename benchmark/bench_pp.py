"""Pipeline-parallel p2p benchmark: credit-based SPSC ring send/recv
latency + bandwidth between adjacent ranks (reference parity:
python/triton_dist/benchmark/bench_pp.py — behavior only).

Run: bash scripts/launch.sh 2 benchmark/bench_pp.py [--check]
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist

import triton_dist_amd as td
from triton_dist_amd.ops import create_p2p_context, p2p_recv, p2p_send


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--check", action="store_true")
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    td.initialize_distributed()
    heap = td.init_symm_heap(size_mb=512)
    world, rank = heap.world, heap.rank
    assert world >= 2, "bench_pp needs >= 2 ranks"
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    nxt, prv = (rank + 1) % world, (rank - 1) % world

    for nbytes in (4 << 10, 1 << 20, 16 << 20):
        elems = nbytes // 2
        ctx = create_p2p_context(max_bytes=nbytes, depth=4)
        x = torch.arange(elems, device=dev).remainder(97).to(torch.bfloat16)
        y = torch.empty_like(x)
        dist.barrier()
        t0 = time.perf_counter()
        for it in range(args.iters):
            if rank % 2 == 0:
                p2p_send(x, nxt, ctx)
                p2p_recv(y, prv, ctx)
            else:
                p2p_recv(y, prv, ctx)
                p2p_send(x, nxt, ctx)
        if dev == "cuda":
            torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / args.iters * 1e6
        if args.check:
            exp = torch.arange(elems).remainder(97).to(torch.bfloat16)
            assert torch.equal(y.cpu(), exp), (rank, nbytes)
        if rank == 0:
            print(f"p2p ring {nbytes/1e6:7.2f} MB: {us:8.1f} us/exchange "
                  f"({nbytes/us/1e3:6.2f} GB/s)")
    if rank == 0 and args.check:
        print("check OK")
    td.shutdown_heap()


if __name__ == "__main__":
    main()
