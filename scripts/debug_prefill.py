"""Minimal repro for the fused prefill prologue + FA2-from-cache path."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from triton_dist_amd import _C  # noqa
from triton_dist_amd.ops.fused import (flash_prefill_op,
                                       qkv_prologue_prefill_op)

def main():
    torch.manual_seed(0)
    b, s, qh, kvh, d = 4, 64, 8, 2, 128
    max_len = 128
    nh = qh + 2 * kvh
    qkv = (torch.randn(b * s, nh * d, device="cuda") / 4).to(torch.bfloat16)
    kc = torch.zeros(b, max_len, kvh, d, dtype=torch.bfloat16, device="cuda")
    vc = torch.zeros(b, max_len, kvh, d, dtype=torch.bfloat16, device="cuda")
    maxp = 256
    inv = 1.0 / (10000 ** (torch.arange(0, 64, device="cuda") / 64.0))
    t = torch.arange(maxp, device="cuda").float()
    ang = torch.outer(t, inv)
    cos, sin = ang.cos().contiguous(), ang.sin().contiguous()
    qnw = torch.ones(d, dtype=torch.bfloat16, device="cuda")
    knw = torch.ones(d, dtype=torch.bfloat16, device="cuda")
    print("launch prologue...", flush=True)
    q4 = qkv_prologue_prefill_op(qkv, kc, vc, cos, sin, qnw, knw, b, s,
                                 qh, kvh, 1e-6, True)
    torch.cuda.synchronize()
    print("prologue ok", q4.shape, flush=True)

    # reference
    qkv3 = qkv.view(b, s, nh, d).float()
    q = qkv3[:, :, :qh]
    k = qkv3[:, :, qh:qh + kvh]
    v = qkv3[:, :, qh + kvh:]
    def rms(x, w):
        return x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()
    q = rms(q, qnw); k = rms(k, knw)
    def rope(x):
        x1, x2 = x[..., :64], x[..., 64:]
        c = cos[:s].view(1, s, 1, 64); si = sin[:s].view(1, s, 1, 64)
        return torch.cat([x1 * c - x2 * si, x2 * c + x1 * si], -1)
    qr, kr = rope(q), rope(k)
    err_q = (q4.float() - qr).abs().max().item()
    err_k = (kc[:, :s].float() - kr).abs().max().item()
    err_v = (vc[:, :s].float() - v).abs().max().item()
    print(f"prologue errs q={err_q:.3e} k={err_k:.3e} v={err_v:.3e}", flush=True)

    print("launch fa2-from-cache...", flush=True)
    kb = max_len * kvh * d
    o = flash_prefill_op(q4, kc, vc, causal=True, kb_stride=kb)
    torch.cuda.synchronize()
    print("fa2 ok", o.shape, flush=True)
    import torch.nn.functional as F
    ref = F.scaled_dot_product_attention(
        qr.permute(0, 2, 1, 3), kr.permute(0, 2, 1, 3),
        v.permute(0, 2, 1, 3), is_causal=True,
        enable_gqa=True).permute(0, 2, 1, 3)
    print("attn err", (o.float() - ref).abs().max().item(), flush=True)


if __name__ == "__main__":
    main()
