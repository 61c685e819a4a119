"""Test helpers: deterministic tensors + tolerance checks.

Counterpart of the reference's test utilities (Triton-distributed
python/triton_dist/test/utils.py assert_allclose, utils.py:986 rand_tensor).
"""
from __future__ import annotations

import torch


def rand_tensor(shape, dtype=torch.bfloat16, device="cpu", scale=1.0,
                generator=None):
    if dtype in (torch.int8, torch.int32, torch.int64):
        return torch.randint(-64, 64, shape, dtype=dtype, device=device,
                             generator=generator)
    t = torch.randn(shape, dtype=torch.float32, device=device,
                    generator=generator) * scale
    return t.to(dtype)


def assert_allclose(actual: torch.Tensor, expected: torch.Tensor,
                    atol=2e-2, rtol=2e-2, max_mismatch_ratio=0.0, msg=""):
    actual32 = actual.float()
    expected32 = expected.float()
    if max_mismatch_ratio > 0:
        diff = (actual32 - expected32).abs()
        tol = atol + rtol * expected32.abs()
        bad = (diff > tol).float().mean().item()
        if bad > max_mismatch_ratio:
            raise AssertionError(
                f"{msg} mismatch ratio {bad:.4f} > {max_mismatch_ratio} "
                f"(max diff {diff.max().item():.4f})")
        return
    torch.testing.assert_close(actual32, expected32, atol=atol, rtol=rtol,
                               msg=msg or None)


def bf16_gemm_tol(k: int):
    """Tolerance for a bf16 GEMM with fp32 accumulation vs an fp32 reference:
    error grows ~sqrt(K) * eps_bf16 * |value|."""
    import math

    rtol = max(2e-2, 0.8e-2 * math.sqrt(k / 1024.0))
    return dict(atol=0.1, rtol=rtol)
