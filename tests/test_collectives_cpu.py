"""CPU (gloo) tests for the standalone collectives: reduce_scatter and
ll_all_gather semantics vs single-process golden references."""
import torch
import torch.distributed as dist

from tests.conftest import run_distributed


def _worker_rs(rank, world):
    import triton_dist_amd as td
    from triton_dist_amd.ops import (all_to_all_single, create_coll_context,
                                     reduce_scatter, ll_all_gather)

    td.init_symm_heap(size_mb=32)
    ctx = create_coll_context(max_seg_elems=4096, max_ll_words=1024)

    g = torch.Generator().manual_seed(7)
    xs = [torch.randn(world * 16, 32, generator=g).to(torch.bfloat16)
          for _ in range(world)]
    x = xs[rank]
    out = reduce_scatter(x, ctx)
    ref = sum(t.float() for t in xs).reshape(world, 16, 32)[rank]
    assert torch.allclose(out.float(), ref, atol=0.25, rtol=0.05), \
        (out.float() - ref).abs().max()

    y = torch.arange(8, dtype=torch.float32).reshape(2, 4) + rank * 100
    gathered = ll_all_gather(y, ctx)
    for r in range(world):
        exp = torch.arange(8, dtype=torch.float32).reshape(2, 4) + r * 100
        assert torch.equal(gathered[r * 2:(r + 1) * 2], exp)
    z = ((torch.arange(world * 16 * 8, dtype=torch.float32) % 97)
         .reshape(world * 16, 8).to(torch.bfloat16) + rank)
    a2a = all_to_all_single(z, ctx)
    for p in range(world):
        exp = z.float().reshape(world, 16, 8)[rank] + (p - rank)
        got = a2a.float().reshape(world, 16, 8)[p]
        assert torch.equal(got, exp), (rank, p)
    td.shutdown_heap()


def test_reduce_scatter_ll_allgather_2rank():
    run_distributed(_worker_rs, world_size=2)


def test_collectives_4rank():
    run_distributed(_worker_rs, world_size=4)


def _worker_a2a_gemm(rank, world):
    import triton_dist_amd as td
    from triton_dist_amd.ops import a2a_gemm, create_ag_gemm_context

    td.init_symm_heap(size_mb=32)
    K, N, M = 32, 24, 8
    ctx = create_ag_gemm_context(max_m_per_rank=M, k=K)
    g = torch.Generator().manual_seed(5)
    xs = [(torch.randn(world * M, K, generator=g) * 0.5).to(torch.bfloat16)
          for _ in range(world)]
    w = (torch.randn(N, K, generator=g) * 0.5).to(torch.bfloat16)
    out = a2a_gemm(xs[rank], w, ctx)
    mixed = torch.cat([xs[s].reshape(world, M, K)[rank].float()
                       for s in range(world)])
    ref = mixed @ w.float().t()
    assert (out.float() - ref).abs().max() < 0.3
    td.shutdown_heap()


def test_a2a_gemm_2rank():
    run_distributed(_worker_a2a_gemm, world_size=2)
