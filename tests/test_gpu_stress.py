"""Randomized-shape GPU stress tests for the fused comm ops (2 ranks over
hipIpc). Mirrors the reference's stress strategy
(test/stress/stress_test_ag_gemm.py, test/amd/test_ag_gemm_intra_node.py:
208-260 — behavior only): many rounds of random shapes through ONE
context, checking against torch.distributed goldens every round, so
flag-reuse/reset bugs and shape-edge bugs surface."""
import random

import pytest
import torch

from tests.conftest import run_distributed

pytestmark = pytest.mark.gpu

ROUNDS = 8


def _body_stress_ag(rank, world):
    from triton_dist_amd.ops import ag_gemm, ag_gemm_ref, create_ag_gemm_context
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    max_m, max_k = 1024, 2048
    ctx = create_ag_gemm_context(max_m_per_rank=max_m, k=max_k,
                                 chunks_per_rank=4)
    rng = random.Random(1234)  # same shape stream on every rank
    for rnd in range(ROUNDS):
        m = rng.choice([128, 256, 512, 768, 1024])
        n = rng.choice([256, 512, 768, 1280])
        method = rng.choice(["push", "fused"])
        if method == "fused" and (m % 256 or n % 256):
            method = "push"
        torch.manual_seed(rnd * 17 + rank)
        a = (torch.randn(m, max_k, device="cuda") / 8).to(torch.bfloat16)
        torch.manual_seed(rnd * 31)
        w = (torch.randn(n, max_k, device="cuda") / 8).to(torch.bfloat16)
        c = ag_gemm(a, w, ctx, method=method)
        torch.cuda.synchronize()
        ref = ag_gemm_ref(a, w)
        assert_allclose(c, ref, msg=f"round {rnd} {method} m={m} n={n}",
                        **bf16_gemm_tol(max_k))


def test_stress_ag_gemm_2rank():
    run_distributed(_body_stress_ag, world_size=2)


def _body_stress_gemm_rs(rank, world):
    from triton_dist_amd.ops import create_gemm_rs_context, gemm_rs, gemm_rs_ref
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    n = 1024
    ctx = create_gemm_rs_context(max_m_total=4096, n=n)
    rng = random.Random(77)
    for rnd in range(ROUNDS):
        m = rng.choice([256, 512, 1024, 2048, 4096])
        k = rng.choice([512, 1024, 1536])
        torch.manual_seed(rnd * 13 + rank)
        a = (torch.randn(m, k, device="cuda") / 8).to(torch.bfloat16)
        torch.manual_seed(rnd * 7)
        w = (torch.randn(n, k, device="cuda") / 8).to(torch.bfloat16)
        c = gemm_rs(a, w, ctx)
        torch.cuda.synchronize()
        ref = gemm_rs_ref(a, w)
        assert_allclose(c, ref, msg=f"round {rnd} m={m} k={k}",
                        atol=2.5e-1, rtol=5e-2)


def test_stress_gemm_rs_2rank():
    run_distributed(_body_stress_gemm_rs, world_size=2)


def _body_stress_gemm_ar(rank, world):
    import torch.distributed as dist
    from triton_dist_amd.ops import create_allreduce_context, gemm_allreduce
    from triton_dist_amd.utils.testing import assert_allclose

    ctx = create_allreduce_context(max_elems=2048 * 2048)
    rng = random.Random(4321)
    for rnd in range(ROUNDS):
        m = rng.choice([256, 512, 1024, 2048])
        n = rng.choice([256, 512, 1024, 2048])
        k = rng.choice([128, 384, 512, 1024])
        if m * n > ctx.max_elems:
            n = ctx.max_elems // m
        torch.manual_seed(rnd * 11 + rank)
        a = (torch.randn(m, k, device="cuda") / 8).to(torch.bfloat16)
        w = (torch.randn(n, k, device="cuda") / 8).to(torch.bfloat16)
        c = gemm_allreduce(a, w, ctx)
        torch.cuda.synchronize()
        ref = a.float() @ w.float().t()
        dist.all_reduce(ref)
        assert_allclose(c, ref.to(torch.bfloat16),
                        msg=f"round {rnd} {m}x{n}x{k}", atol=2.5e-1,
                        rtol=5e-2)


def test_stress_gemm_ar_2rank():
    run_distributed(_body_stress_gemm_ar, world_size=2)
