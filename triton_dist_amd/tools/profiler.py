"""Host side of the intra-kernel profiler + timing utilities.

Capability parity with Triton-distributed tools/profiler/{context.py,
viewer.py} (ProfilerBuffer context manager + perfetto/chrome export) and
profiler_utils.py:205-303 group_profile (merged multi-rank traces).
"""
from __future__ import annotations

import json
from pathlib import Path
from typing import Optional

import torch

TAG_NAMES = {0: "tile_wait", 1: "tile_compute"}
WALLCLOCK_HZ = 100e6  # gfx950 s_memrealtime


class KernelProfiler:
    """Device ring buffer for kprof_record (csrc/include/td/profiler.hpp).

    Usage:
        prof = KernelProfiler(capacity=1 << 16)
        ag_gemm(..., profiler=prof)         # op passes buf/cursor pointers
        prof.export_chrome_trace("trace.json", rank=rank)
    """

    def __init__(self, capacity: int = 1 << 16, device="cuda"):
        self.capacity = capacity
        self.buf = torch.zeros(capacity, 4, dtype=torch.int64, device=device)
        self.cursor = torch.zeros(1, dtype=torch.int32, device=device)

    def ptrs(self):
        return self.buf.data_ptr(), self.cursor.data_ptr(), self.capacity

    def reset(self):
        self.cursor.zero_()

    def records(self):
        n = min(int(self.cursor.item()), self.capacity)
        return self.buf[:n].cpu()

    def export_chrome_trace(self, path: str, rank: int = 0,
                            merge_into: Optional[list] = None):
        recs = self.records()
        events = merge_into if merge_into is not None else []
        if len(recs):
            t_base = int(recs[:, 2].min())
            for block, tag, t0, t1 in recs.tolist():
                events.append({
                    "name": TAG_NAMES.get(int(tag), f"tag{tag}"),
                    "ph": "X", "pid": rank, "tid": int(block) & 0xFFFFFFFF,
                    "ts": (t0 - t_base) / WALLCLOCK_HZ * 1e6,
                    "dur": max(t1 - t0, 0) / WALLCLOCK_HZ * 1e6,
                })
        if merge_into is None:
            Path(path).write_text(json.dumps({"traceEvents": events}))
        return events

    def summary(self):
        """Per-tag total/mean µs — the overlap headline (tile_wait is the
        exposed communication time inside the consumer GEMM)."""
        recs = self.records()
        out = {}
        for tag, name in TAG_NAMES.items():
            sel = recs[recs[:, 1] == tag]
            if len(sel):
                dur = (sel[:, 3] - sel[:, 2]).clamp(min=0).float() \
                    / WALLCLOCK_HZ * 1e6
                out[name] = {"count": len(sel),
                             "total_us": float(dur.sum()),
                             "mean_us": float(dur.mean()),
                             "max_us": float(dur.max())}
        return out
