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


class group_profile:
    """Merged multi-rank torch-profiler traces (capability parity with
    Triton-distributed profiler_utils.py:205-303 group_profile): each rank
    exports chrome traces under <dir>/rank<r>.json; rank 0 merges them into
    one trace with per-rank pids after the barrier."""

    def __init__(self, name="trace", do_prof=True, out_dir="prof_out",
                 group=None):
        self.name, self.do_prof, self.out_dir = name, do_prof, out_dir
        self.group = group
        self.prof = None

    def __enter__(self):
        if self.do_prof:
            import torch.profiler as tp

            acts = [tp.ProfilerActivity.CPU]
            if torch.cuda.is_available():
                acts.append(tp.ProfilerActivity.CUDA)
            self.prof = tp.profile(activities=acts)
            self.prof.__enter__()
        return self

    def __exit__(self, *exc):
        if self.prof is None:
            return False
        import json
        import os

        import torch.distributed as dist

        self.prof.__exit__(*exc)
        rank = dist.get_rank(self.group) if dist.is_initialized() else 0
        world = dist.get_world_size(self.group) if dist.is_initialized() else 1
        os.makedirs(self.out_dir, exist_ok=True)
        path = os.path.join(self.out_dir, f"{self.name}_rank{rank}.json")
        self.prof.export_chrome_trace(path)
        if dist.is_initialized():
            dist.barrier(self.group)
        if rank == 0 and world > 1:
            events = []
            for r in range(world):
                p = os.path.join(self.out_dir, f"{self.name}_rank{r}.json")
                try:
                    data = json.load(open(p))
                    for ev in data.get("traceEvents", []):
                        ev["pid"] = f"rank{r}.{ev.get('pid', 0)}"
                        events.append(ev)
                except Exception:
                    pass
            json.dump({"traceEvents": events},
                      open(os.path.join(self.out_dir,
                                        f"{self.name}_merged.json"), "w"))
        return False


def wait_stable_clock(frac: float = 0.9, timeout_s: float = 5.0) -> bool:
    """Spin a dummy workload until the GPU sclk reaches `frac` of its max
    (capability parity with the reference's GPU-clock wait for stable
    benchmarking, utils.py:953 — behavior only). Returns True when the
    clock stabilized, False on timeout / no clock telemetry."""
    if not torch.cuda.is_available():
        return False
    import re
    import subprocess

    def read_clock():
        try:
            out = subprocess.run(["rocm-smi", "--showgpuclocks"],
                                 capture_output=True, text=True,
                                 timeout=3).stdout
            m = re.findall(r"sclk.*?\((\d+)Mhz\)", out)
            return int(m[0]) if m else None
        except Exception:
            return None

    mx = None
    try:
        out = subprocess.run(["rocm-smi", "-s"], capture_output=True,
                             text=True, timeout=3).stdout
        lv = re.findall(r"(\d+)Mhz", out)
        mx = max(int(v) for v in lv) if lv else None
    except Exception:
        pass
    if mx is None:
        mx = 2400  # MI355X nominal peak
    x = torch.randn(2048, 2048, device="cuda", dtype=torch.bfloat16)
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < timeout_s:
        for _ in range(10):
            x = x @ x * 1e-3
        torch.cuda.synchronize()
        clk = read_clock()
        if clk is not None and clk >= frac * mx:
            return True
    return False
