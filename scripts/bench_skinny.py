"""A/B the skinny-M decode GEMM tier (k_gemm_skinny) against hipBLASLt
and the gemm256_sk split-K tier on the qwen3-32b TP1 decode shapes
(numerics first)."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from triton_dist_amd import _C


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
    shapes = [  # (name, m, n, k) — qwen3-32b TP1 decode
        ("qkv", 512, 10240, 5120),
        ("o", 512, 5120, 8192),
        ("gate_up", 512, 55296, 5120),
        ("down", 512, 5120, 27648),
        ("moe-ish", 512, 1536, 2048),
    ]
    for name, m, n, k in shapes:
        a = (torch.randn(m, k, device="cuda") / 8).to(torch.bfloat16)
        w = (torch.randn(n, k, device="cuda") / 8).to(torch.bfloat16)
        c = torch.empty(m, n, device="cuda", dtype=torch.bfloat16)
        ref = (a.float() @ w.float().t())
        _C.gemm_skinny_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(), 0,
                            m, n, k, 0, s)
        torch.cuda.synchronize()
        rel = ((c.float() - ref).abs().max() / ref.abs().max()).item()
        us_sk = t(lambda: _C.gemm_skinny_bf16(
            a.data_ptr(), w.data_ptr(), c.data_ptr(), 0, m, n, k, 0, s))
        us_blt = t(lambda: torch.matmul(a, w.t(), out=c))
        us_ring = None
        if m % 256 == 0 and n % 256 == 0 and k % 128 == 0:
            us_ring = t(lambda: _C.gemm_bf16(
                a.data_ptr(), w.data_ptr(), c.data_ptr(), 0, m, n, k, s))
        floor = n * k * 2 / 8e12 * 1e6  # weight bytes / 8 TB/s
        line = (f"{name:8s} {m}x{n}x{k}: skinny {us_sk:6.1f} us"
                f" blt {us_blt:6.1f} us floor {floor:5.1f} us"
                f" rel {rel:.1e}")
        if us_ring is not None:
            line += f" ring {us_ring:6.1f} us"
        from triton_dist_amd.ops.gemm import sk256_pick
        skf = sk256_pick(m, n, k)
        if skf:
            ws = torch.empty(m, n, dtype=torch.float32, device="cuda")
            sweep = []
            for sv in sorted({2, 3, 4, 5, skf, 2 * skf}):
                if k % (128 * sv) or (k // 128 // sv) < 2:
                    continue
                us = t(lambda sv=sv: _C.gemm256_sk_bf16(
                    a.data_ptr(), w.data_ptr(), c.data_ptr(), 0,
                    ws.data_ptr(), m, n, k, sv, s))
                ws2 = torch.empty(sv, m, n, dtype=torch.float32,
                                  device="cuda")
                us2 = t(lambda sv=sv, ws2=ws2: _C.gemm256_sk2_bf16(
                    a.data_ptr(), w.data_ptr(), c.data_ptr(), 0,
                    ws2.data_ptr(), m, n, k, sv, s))
                sweep.append(f"sk{sv}:{us:.0f}/{us2:.0f}")
            line += " sk-atomic/2stage[" + " ".join(sweep) + "]us"
        print(line)


if __name__ == "__main__":
    main()
