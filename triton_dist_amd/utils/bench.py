"""Benchmark timing helpers (cf. Triton-distributed
python/triton_dist/profiler_utils.py:330-371 perf_func /
perf_func_with_l2_reset — semantics only).
"""
from __future__ import annotations

import time

import torch


def perf_func(fn, iters=20, warmup=5, sync_all_ranks=True):
    """Time `fn` with CUDA events on GPU (with a short async sleep to decouple
    launch overhead), wall clock on CPU. Returns (last_output, ms_per_iter).
    """
    import torch.distributed as dist

    out = None
    for _ in range(warmup):
        out = fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
        if sync_all_ranks and dist.is_initialized():
            dist.barrier()
        start = torch.cuda.Event(enable_timing=True)
        end = torch.cuda.Event(enable_timing=True)
        torch.cuda._sleep(int(2e6))  # decouple from host launch jitter
        start.record()
        for _ in range(iters):
            out = fn()
        end.record()
        torch.cuda.synchronize()
        ms = start.elapsed_time(end) / iters
    else:
        if sync_all_ranks and dist.is_initialized():
            dist.barrier()
        t0 = time.perf_counter()
        for _ in range(iters):
            out = fn()
        ms = (time.perf_counter() - t0) * 1e3 / iters
    return out, ms


def perf_func_with_l2_reset(fn, iters=20, warmup=5, l2_size_mb=300):
    """Flush L2/L3 between timed iterations (256 MiB Infinity Cache on
    MI355X means small working sets look artificially fast otherwise)."""
    if not torch.cuda.is_available():
        return perf_func(fn, iters, warmup)
    cache = torch.empty(l2_size_mb * 1024 * 1024, dtype=torch.uint8,
                        device="cuda")
    out = None
    for _ in range(warmup):
        out = fn()
    torch.cuda.synchronize()
    total_ms = 0.0
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    for _ in range(iters):
        cache.zero_()
        torch.cuda.synchronize()
        start.record()
        out = fn()
        end.record()
        torch.cuda.synchronize()
        total_ms += start.elapsed_time(end)
    return out, total_ms / iters
