#!/usr/bin/env python3
"""xGMI / heap bandwidth sweep (capability parity with the reference's
test/amd/test_bandwidth.py + memory_ops toolkit): SDMA memcpy (push),
SM copy kernel, and put_signal across peers.
Run: bash scripts/launch.sh 2 benchmark/bench_bandwidth.py"""
import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--sizes", default="65536,1048576,16777216,268435456")
    p.add_argument("--iters", type=int, default=20)
    args = p.parse_args()

    import triton_dist_amd as td
    from triton_dist_amd import _C
    from triton_dist_amd.utils import perf_func

    td.initialize_distributed()
    heap = td.init_symm_heap()
    world, rank = heap.world, heap.rank
    sizes = [int(s) for s in args.sizes.split(",")]
    src = heap.alloc_buffer((max(sizes),), torch.uint8)
    dst = heap.alloc_buffer((max(sizes),), torch.uint8)
    flag = heap.alloc_buffer((8,), torch.int32)
    peer = (rank + 1) % world
    s = torch.cuda.current_stream().cuda_stream
    for n in sizes:
        rows = []
        for name, fn in (
            ("sdma_push", lambda: _C.memcpy_async(dst.ptr(peer), src.ptr(),
                                                  n, s)),
            ("sm_copy", lambda: _C.copy_kernel(dst.ptr(peer), src.ptr(), n,
                                               s)),
            ("put_signal", lambda: _C.put_signal(dst.ptr(peer), src.ptr(),
                                                 n, flag.ptr(peer), 1,
                                                 True, s)),
        ):
            _, ms = perf_func(fn, iters=args.iters, warmup=3,
                              sync_all_ranks=False)
            rows.append(f"{name} {n / ms / 1e6:8.1f} GB/s")
        td.dist_print(f"{n / 1048576:8.2f} MiB -> {' | '.join(rows)}",
                      allowed_ranks="0")
    td.finalize_distributed()


if __name__ == "__main__":
    main()
