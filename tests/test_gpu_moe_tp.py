"""GPU test: TP-MoE path (AG + pq grouped GEMM + topk-reduce + RS) on 2
ranks sharing one GPU via hipIpc, vs the full-weight golden reference."""
import pytest
import torch

from tests.conftest import run_distributed

pytestmark = pytest.mark.gpu

E, K, H, INTER, M_LOC = 16, 4, 256, 256, 128


def _worker(rank, world):
    import triton_dist_amd as td
    from triton_dist_amd.ops import (create_ag_gemm_context,
                                     create_coll_context, tp_moe_forward,
                                     tp_moe_ref)

    td.init_symm_heap(size_mb=256)
    ag_ctx = create_ag_gemm_context(max_m_per_rank=M_LOC, k=H)
    coll_ctx = create_coll_context(max_seg_elems=M_LOC * H,
                                   max_ll_words=256)

    g = torch.Generator().manual_seed(3)
    gate = torch.randn(E, INTER, H, generator=g) * 0.2
    up = torch.randn(E, INTER, H, generator=g) * 0.2
    down = torch.randn(E, H, INTER, generator=g) * 0.2
    inter_s = INTER // world
    lo = rank * inter_s
    w_gate_up = torch.cat(
        [gate[:, lo:lo + inter_s], up[:, lo:lo + inter_s]],
        dim=1).to(torch.bfloat16).cuda().contiguous()
    w_down = down[:, :, lo:lo + inter_s].to(
        torch.bfloat16).cuda().contiguous()

    g2 = torch.Generator().manual_seed(17)
    x_full = (torch.randn(world * M_LOC, H, generator=g2) * 0.5).to(
        torch.bfloat16)
    ids = torch.randint(0, E, (world * M_LOC, K), generator=g2,
                        dtype=torch.int32)
    w = torch.softmax(torch.randn(world * M_LOC, K, generator=g2), dim=-1)

    x_shard = x_full[rank * M_LOC:(rank + 1) * M_LOC].cuda().contiguous()
    ref = tp_moe_ref(x_full, ids, w,
                     torch.cat([gate, up], dim=1).to(torch.bfloat16),
                     down.to(torch.bfloat16), world, rank)
    # both paths: gather-then-compute and the segment-progressive
    # AG<->grouped-GEMM overlap (rank-staggered arrival order)
    for overlap in (False, True):
        out = tp_moe_forward(x_shard, ids.cuda(), w.cuda(), w_gate_up,
                             w_down, ag_ctx, coll_ctx, overlap=overlap)
        torch.cuda.synchronize()
        err = (out.float().cpu() - ref.float()).abs().max().item()
        rel = err / ref.float().abs().max().item()
        assert rel < 0.05, (rank, overlap, err, rel)
    td.shutdown_heap()


def test_tp_moe_gpu_2rank():
    run_distributed(_worker, world_size=2)
