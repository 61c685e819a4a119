"""Numerics + A/B for the weight-streaming decode GEMM (gemm_stream.hip)
vs hipBLASLt and the current best_gemm route, on the TP1 decode shapes
of qwen3-32b / seed-oss-36b. Sweeps sk per shape."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from triton_dist_amd import _C
from triton_dist_amd.ops.gemm import best_gemm


def t(fn, n=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6


def main():
    s = torch.cuda.current_stream().cuda_stream
    shapes = [  # (name, m, n, k)
        ("qkv32b", 512, 10240, 5120),
        ("o32b", 512, 5120, 8192),
        ("gup32b", 512, 51200, 5120),
        ("down32b", 512, 5120, 25600),
        ("gup36b", 512, 55296, 5120),
        ("down36b", 512, 5120, 27648),
    ]
    for name, m, n, k in shapes:
        a = (torch.randn(m, k, device="cuda") / 8).to(torch.bfloat16)
        w = (torch.randn(n, k, device="cuda") / 8).to(torch.bfloat16)
        c = torch.empty(m, n, device="cuda", dtype=torch.bfloat16)
        ref = a.float() @ w.float().t()
        floor = n * k * 2 / 8e12 * 1e6
        us_blt = t(lambda: torch.matmul(a, w.t(), out=c))
        us_best = t(lambda: best_gemm(a, w, out=c))
        line = (f"{name:8s} {m}x{n}x{k}: blt {us_blt:6.1f} best "
                f"{us_best:6.1f} floor {floor:5.1f} | stream")
        for sk in (1, 2, 4, 5, 8):
            if k % (32 * sk) or (k // 32) // sk < 2:
                continue
            ws = torch.empty(sk, m, n, dtype=torch.float32,
                             device="cuda") if sk > 1 else c
            _C.gemm_stream_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(),
                                0, ws.data_ptr(), m, n, k, sk, s)
            torch.cuda.synchronize()
            rel = ((c.float() - ref).abs().max()
                   / ref.abs().max()).item()
            us = t(lambda: _C.gemm_stream_bf16(
                a.data_ptr(), w.data_ptr(), c.data_ptr(), 0,
                ws.data_ptr(), m, n, k, sk, s))
            ok = "" if rel < 3e-2 else f"!REL{rel:.0e}"
            line += f" sk{sk}:{us:5.1f}{ok}"
        print(line, flush=True)


if __name__ == "__main__":
    main()
