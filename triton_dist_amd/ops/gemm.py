"""Plain persistent MFMA GEMM wrapper: C = A @ W^T (+bias), bf16.

The HIP kernel (csrc/kernels/gemm.hip k_gemm_bf16) replaces the reference's
Triton GEMM zoo (Triton-distributed python/triton_dist/kernels/amd/gemm.py:
62-541 — capability only). Weights are stored [N, K] (torch.nn.Linear
layout), which makes A and W fragment loads symmetric on MFMA.
"""
from __future__ import annotations

import torch


def _native():
    from .. import _C

    if _C is None:
        raise RuntimeError("triton_dist_amd._C not built (required on GPU)")
    return _C


def gemm_supported(m: int, n: int, k: int) -> bool:
    return m % 128 == 0 and n % 128 == 0 and k % 64 == 0


def choose_splits(m: int, n: int, k: int) -> int:
    """Split-K factor for occupancy-starved shapes: the 128-tile grid must
    reach ~200 workgroups (256 CUs) before split-K stops paying."""
    grid = (m // 128) * (n // 128)
    if grid >= 208:
        return 1
    for s in (8, 4, 2):
        if (k // 32) % s == 0 and k // s >= 512 and grid * s <= 2048:
            return s
    return 1


def splitk_ws(m: int, n: int, splits: int, device) -> torch.Tensor:
    return torch.empty(splits, m, n, dtype=torch.float32, device=device)


def sk256_pick(m: int, n: int, k: int) -> int:
    """Split factor for the 256^2 fp32-atomic split-K tier, or 0 if the
    shape doesn't profit. Measured (M=512 decode shapes, MI355X): wins only
    in the K >> N regime (down-proj: 188 us vs hipBLASLt 262); loses when
    N is large (atomic+convert traffic) or K is short (pipeline prologue).
    Target ~200 workgroups (256 CUs, 1 wg/CU at 128 KiB LDS)."""
    if m % 256 or n % 256 or k < 4 * n or k < 8192:
        return 0
    grid = (m // 256) * (n // 256)
    best, best_d = 0, 1 << 30
    for s in (2, 4, 5, 8, 10, 16, 20, 25):
        if k % (128 * s) or (k // 128 // s) < 4:
            continue
        d = abs(grid * s - 200)
        if d < best_d:
            best, best_d = s, d
    return best


def best_gemm(a: torch.Tensor, w: torch.Tensor,
              bias: torch.Tensor | None = None,
              out: torch.Tensor | None = None) -> torch.Tensor:
    """C = A @ W^T + bias via the fastest available plain-GEMM backend:
    hipBLASLt (torch.matmul) in general, the in-house gemm256_sk tier for
    K>>N decode shapes where it measures faster."""
    m, k = a.shape
    n = w.shape[0]
    if a.is_cuda and a.dtype == torch.bfloat16:
        sk = sk256_pick(m, n, k)
        if sk:
            if out is None:
                out = torch.empty(m, n, dtype=torch.bfloat16,
                                  device=a.device)
            # two-stage split-K (private fp32 slices + fused reduce):
            # measured 213 vs 228 us on the down-proj shape vs the
            # atomic tier (profiles/README.md r02)
            ws = torch.empty(sk, m, n, dtype=torch.float32,
                             device=a.device)
            _native().gemm256_sk2_bf16(
                a.data_ptr(), w.data_ptr(), out.data_ptr(),
                bias.data_ptr() if bias is not None else 0,
                ws.data_ptr(), m, n, k, sk,
                torch.cuda.current_stream().cuda_stream)
            return out
    c = torch.matmul(a, w.t(), out=out) if out is not None else a @ w.t()
    if bias is not None:
        c += bias
    return c


def gemm(a: torch.Tensor, w: torch.Tensor, bias: torch.Tensor | None = None,
         out: torch.Tensor | None = None) -> torch.Tensor:
    """C[M,N] = A[M,K] @ W[N,K]^T + bias."""
    m, k = a.shape
    n, k2 = w.shape
    assert k == k2, (a.shape, w.shape)
    if a.is_cuda:
        assert a.dtype == torch.bfloat16 and w.dtype == torch.bfloat16
        assert a.is_contiguous() and w.is_contiguous()
        assert gemm_supported(m, n, k), \
            f"gemm v1 needs M%128==0,N%128==0,K%64==0, got {(m, n, k)}"
        if out is None:
            out = torch.empty(m, n, dtype=torch.bfloat16, device=a.device)
        stream = torch.cuda.current_stream().cuda_stream
        splits = choose_splits(m, n, k)
        if splits > 1:
            ws = splitk_ws(m, n, splits, a.device)
            _native().gemm_splitk_bf16(
                a.data_ptr(), w.data_ptr(), out.data_ptr(),
                bias.data_ptr() if bias is not None else 0,
                ws.data_ptr(), m, n, k, splits, stream)
        else:
            _native().gemm_bf16(
                a.data_ptr(), w.data_ptr(), out.data_ptr(),
                bias.data_ptr() if bias is not None else 0,
                m, n, k, stream)
        return out
    # CPU reference path
    c = a.float() @ w.float().t()
    if bias is not None:
        c += bias.float()
    c = c.to(a.dtype)
    if out is not None:
        out.copy_(c)
        return out
    return c


def gemm_ref(a: torch.Tensor, w: torch.Tensor,
             bias: torch.Tensor | None = None) -> torch.Tensor:
    """fp32 golden reference."""
    c = a.float() @ w.float().t()
    if bias is not None:
        c += bias.float()
    return c
