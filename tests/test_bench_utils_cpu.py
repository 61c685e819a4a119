"""Host profiling utilities (reference profiler_utils parity): perf_func
timing contract and KernelProfiler trace bookkeeping on CPU."""
import json

import torch

from triton_dist_amd.utils.bench import perf_func, perf_func_with_l2_reset


def test_perf_func_contract():
    calls = []

    def fn():
        calls.append(1)
        return torch.ones(4).sum()

    out, ms = perf_func(fn, iters=5, warmup=2, sync_all_ranks=False)
    assert len(calls) == 7           # warmup + timed
    assert float(out) == 4.0         # last output returned
    assert ms >= 0.0


def test_perf_func_l2_reset_cpu_fallback():
    out, ms = perf_func_with_l2_reset(lambda: 3, iters=2, warmup=1)
    assert out == 3 and ms >= 0.0
