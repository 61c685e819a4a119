"""A/B the EXPERIMENTAL BK=64 quadrant-phase GEMM (gemm256_v2) against
the production K-slice ring and hipBLASLt. Round-2 iteration driver —
includes a numerics check first (the v2 ledger/swizzle are unproven on
hardware until this passes)."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from triton_dist_amd import _C


def t(fn, n=30):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6


def main():
    s = torch.cuda.current_stream().cuda_stream
    for m, n, k in [(512, 512, 512), (4096, 4096, 4096),
                    (8192, 8192, 8192)]:
        a = torch.randn(m, k, device="cuda").to(torch.bfloat16) / 8
        w = torch.randn(n, k, device="cuda").to(torch.bfloat16) / 8
        c = torch.empty(m, n, device="cuda", dtype=torch.bfloat16)
        ref = a.float() @ w.float().t()
        _C.gemm256_v2_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(),
                           m, n, k, s)
        torch.cuda.synchronize()
        c3 = torch.empty_like(c)
        _C.gemm256_v3_bf16(a.data_ptr(), w.data_ptr(), c3.data_ptr(),
                           m, n, k, s)
        torch.cuda.synchronize()
        rel3 = ((c3.float() - ref).abs().max() / ref.abs().max()).item()
        rel = ((c.float() - ref).abs().max() / ref.abs().max()).item()
        status = "OK" if rel < 2e-2 else "FAIL"
        status3 = "OK" if rel3 < 2e-2 else "FAIL"
        line = (f"{m}x{n}x{k}: v2 {status} (rel {rel:.1e})"
                f" v3 {status3} (rel {rel3:.1e})")
        if status == "OK" and m >= 4096:
            gf = 2 * m * n * k / 1e9
            us_v2 = t(lambda: _C.gemm256_v2_bf16(
                a.data_ptr(), w.data_ptr(), c.data_ptr(), m, n, k, s))
            us_v3 = t(lambda: _C.gemm256_v3_bf16(
                a.data_ptr(), w.data_ptr(), c.data_ptr(), m, n, k, s))
            us_v1 = t(lambda: _C.gemm_bf16(
                a.data_ptr(), w.data_ptr(), c.data_ptr(), 0, m, n, k, s))
            us_blt = t(lambda: torch.matmul(a, w.t()))
            line += (f" | v2 {gf/us_v2*1e3:5.0f} TF"
                     f" v3 {gf/us_v3*1e3:5.0f} TF"
                     f" ring {gf/us_v1*1e3:5.0f} TF"
                     f" blt {gf/us_blt*1e3:5.0f} TF")
        print(line)


if __name__ == "__main__":
    main()
