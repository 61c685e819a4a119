"""GDN forward benchmarks: chunked prefill (batched GEMM chain) and the
HIP decode-step kernel (state-bandwidth bound).

Run: python benchmark/bench_gdn.py
"""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.nn.functional as F

from triton_dist_amd.ops import chunk_gated_delta_rule_fwd, gdn_decode_step


def main():
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--check", action="store_true")
    args = ap.parse_args()
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    B, T, H, K, V = (1, 8192, 16, 128, 128) if dev == "cuda" \
        else (1, 256, 4, 32, 32)
    q = torch.randn(B, T, H, K, device=dev).to(torch.bfloat16)
    k = F.normalize(torch.randn(B, T, H, K, device=dev), p=2,
                    dim=-1).to(torch.bfloat16)
    v = torch.randn(B, T, H, V, device=dev).to(torch.bfloat16)
    beta = torch.rand(B, T, H, device=dev)
    g = F.logsigmoid(torch.rand(B, T, H, device=dev))
    scale = K ** -0.5

    def sync():
        if dev == "cuda":
            torch.cuda.synchronize()

    if args.check:
        from triton_dist_amd.ops import gated_delta_rule_recurrent_ref

        Bs, Ts = 1, 96
        o_ref, _ = gated_delta_rule_recurrent_ref(
            q[:Bs, :Ts].float().cpu(), k[:Bs, :Ts].float().cpu(),
            v[:Bs, :Ts].float().cpu(), g[:Bs, :Ts].cpu(),
            beta[:Bs, :Ts].cpu(), scale)
        o_c, _ = chunk_gated_delta_rule_fwd(q[:Bs, :Ts], k[:Bs, :Ts],
                                            v[:Bs, :Ts], g[:Bs, :Ts],
                                            beta[:Bs, :Ts], scale)
        rel = ((o_c.float().cpu() - o_ref).abs().max()
               / o_ref.abs().max()).item()
        print(f"check vs recurrent ref: rel {rel:.2e} "
              f"({'OK' if rel < 0.05 else 'FAIL'})")
    for _ in range(2):
        chunk_gated_delta_rule_fwd(q, k, v, g, beta, scale)
    sync()
    t0 = time.perf_counter()
    iters = 5
    for _ in range(iters):
        chunk_gated_delta_rule_fwd(q, k, v, g, beta, scale)
    sync()
    ms = (time.perf_counter() - t0) / iters * 1e3
    print(f"chunked prefill B{B} T{T} H{H} K{K} V{V}: {ms:.2f} ms "
          f"({B * T * H / ms * 1e3 / 1e6:.2f} M tok-heads/s)")

    Bd = 64 if dev == "cuda" else 4
    state = torch.zeros(Bd, H, K, V, dtype=torch.float32, device=dev)
    qd = torch.randn(Bd, H, K, device=dev).to(torch.bfloat16)
    kd = torch.randn(Bd, H, K, device=dev).to(torch.bfloat16)
    vd = torch.randn(Bd, H, V, device=dev).to(torch.bfloat16)
    gd = F.logsigmoid(torch.rand(Bd, H, device=dev))
    bd = torch.rand(Bd, H, device=dev)
    for _ in range(3):
        gdn_decode_step(qd, kd, vd, gd, bd, scale, state)
    sync()
    t0 = time.perf_counter()
    for _ in range(50):
        gdn_decode_step(qd, kd, vd, gd, bd, scale, state)
    sync()
    us = (time.perf_counter() - t0) / 50 * 1e6
    traffic = Bd * H * K * V * 4 * 3  # state r+r+w
    print(f"decode step B{Bd} H{H}: {us:.1f} us "
          f"({traffic / us / 1e3:.2f} GB/s state traffic)")


if __name__ == "__main__":
    main()
