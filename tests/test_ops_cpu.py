"""CPU (gloo + shm mock) tests of the fused-op context/flag logic against
torch.distributed golden references — the reference repo's --check pattern
(test/amd/test_ag_gemm_intra_node.py:53-74 semantics) run on CPU."""
import torch

from tests.conftest import run_distributed


def _body_ag_gemm(rank, world):
    from triton_dist_amd.ops import ag_gemm, ag_gemm_ref, create_ag_gemm_context
    from triton_dist_amd.utils import assert_allclose, rand_tensor

    m, k, n = 64, 32, 48
    ctx = create_ag_gemm_context(max_m_per_rank=m, k=k, chunks_per_rank=4)
    g = torch.Generator().manual_seed(7 + rank)
    a = rand_tensor((m, k), dtype=torch.bfloat16, generator=g)
    gw = torch.Generator().manual_seed(99)  # same weights on all ranks
    w = rand_tensor((n, k), dtype=torch.bfloat16, generator=gw)
    for _ in range(3):  # epoch reuse
        c = ag_gemm(a, w, ctx)
        ref = ag_gemm_ref(a, w)
        assert_allclose(c, ref, atol=5e-2, rtol=5e-2)


def test_ag_gemm_cpu_2rank():
    run_distributed(_body_ag_gemm, world_size=2)


def test_ag_gemm_cpu_4rank():
    run_distributed(_body_ag_gemm, world_size=4)


def _body_gemm_rs(rank, world):
    from triton_dist_amd.ops import (create_gemm_rs_context, gemm_rs,
                                     gemm_rs_ref)
    from triton_dist_amd.utils import assert_allclose, rand_tensor

    m_total, k, n = 64 * world, 32, 48
    ctx = create_gemm_rs_context(max_m_total=m_total, n=n)
    g = torch.Generator().manual_seed(11 + rank)
    a = rand_tensor((m_total, k), dtype=torch.bfloat16, generator=g)
    w = rand_tensor((n, k), dtype=torch.bfloat16,
                    generator=torch.Generator().manual_seed(5))
    for _ in range(3):
        out = gemm_rs(a, w, ctx)
        ref = gemm_rs_ref(a, w)
        assert_allclose(out, ref, atol=8e-2, rtol=8e-2)


def test_gemm_rs_cpu_2rank():
    run_distributed(_body_gemm_rs, world_size=2)


def test_gemm_rs_cpu_4rank():
    run_distributed(_body_gemm_rs, world_size=4)


def test_gemm_cpu_fallback():
    from triton_dist_amd.ops import gemm, gemm_ref
    from triton_dist_amd.utils import assert_allclose, rand_tensor

    a = rand_tensor((64, 32), dtype=torch.bfloat16)
    w = rand_tensor((48, 32), dtype=torch.bfloat16)
    assert_allclose(gemm(a, w), gemm_ref(a, w), atol=5e-2, rtol=5e-2)


def _body_allgather(rank, world):
    from triton_dist_amd.ops import create_ag_gemm_context
    from triton_dist_amd.ops.allgather_gemm import allgather
    from triton_dist_amd.utils import rand_tensor

    m, k = 32, 16
    ctx = create_ag_gemm_context(max_m_per_rank=m, k=k, chunks_per_rank=4)
    for mm in (m, m // 2):  # full + variable m
        g = torch.Generator().manual_seed(3 + rank)
        a = rand_tensor((mm, k), dtype=torch.bfloat16, generator=g)
        out = allgather(a, ctx)
        for r in range(world):
            gr = torch.Generator().manual_seed(3 + r)
            exp = rand_tensor((mm, k), dtype=torch.bfloat16, generator=gr)
            assert torch.equal(out[r * mm:(r + 1) * mm], exp), (rank, r, mm)


def test_allgather_cpu_2rank():
    run_distributed(_body_allgather, world_size=2)


def test_allgather_cpu_4rank():
    run_distributed(_body_allgather, world_size=4)


def test_gpu_oversubscribed_helper(monkeypatch):
    """No GPU here -> always False; with a fake 1-GPU host, a 2-rank
    world reports oversubscribed (the starvation-fallback trigger)."""
    from triton_dist_amd.utils import distributed as d

    assert d.gpu_oversubscribed() is False
    assert d.gpu_oversubscribed(world=8) is False
    monkeypatch.setattr(d, "has_gpu", lambda: True)
    monkeypatch.setattr(d.torch.cuda, "device_count", lambda: 1)
    assert d.gpu_oversubscribed(world=2) is True
    assert d.gpu_oversubscribed(world=1) is False
    monkeypatch.setattr(d.torch.cuda, "device_count", lambda: 8)
    assert d.gpu_oversubscribed(world=8) is False
