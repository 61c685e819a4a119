"""A/B the MFMA FA2 prefill kernel (k_flash_prefill) against torch sdpa
(ROCm aotriton flash) on prefill shapes, with numerics."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.nn.functional as F

from triton_dist_amd.ops.fused import flash_prefill_op


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
    for b, s, qh, kvh in [(32, 128, 64, 8), (512, 128, 64, 8),
                          (2, 2048, 64, 8), (1, 8192, 32, 8)]:
        d = 128
        q = (torch.randn(b, s, qh, d, device="cuda") / 4).to(torch.bfloat16)
        k = (torch.randn(b, s, kvh, d, device="cuda") / 4).to(torch.bfloat16)
        v = (torch.randn(b, s, kvh, d, device="cuda") / 4).to(torch.bfloat16)
        out = flash_prefill_op(q, k, v, causal=True)
        torch.cuda.synchronize()
        qt = q.permute(0, 2, 1, 3).contiguous()
        kt = k.permute(0, 2, 1, 3).contiguous()
        vt = v.permute(0, 2, 1, 3).contiguous()
        ref = F.scaled_dot_product_attention(
            qt.float(), kt.float(), vt.float(), is_causal=True,
            enable_gqa=True).permute(0, 2, 1, 3)
        rel = ((out.float() - ref).abs().max() / ref.abs().max()).item()
        us_fa = t(lambda: flash_prefill_op(q, k, v, causal=True))
        us_sdpa = t(lambda: F.scaled_dot_product_attention(
            qt, kt, vt, is_causal=True, enable_gqa=True))
        print(f"b={b} s={s} qh={qh}/{kvh}: fa2 {us_fa:8.1f} us"
              f"  sdpa {us_sdpa:8.1f} us  rel {rel:.1e}")


if __name__ == "__main__":
    main()
