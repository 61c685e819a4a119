"""GPU tests for the standalone collectives (2 ranks sharing one GPU via
hipIpc): reduce_scatter push+reduce protocol and the LL flag-in-payload
allgather, against torch references."""
import pytest
import torch

from tests.conftest import run_distributed

pytestmark = pytest.mark.gpu


def _body_collectives(rank, world):
    import triton_dist_amd as td
    from triton_dist_amd.ops import (all_to_all_single, create_coll_context,
                                     reduce_scatter, ll_all_gather)

    td.init_symm_heap(size_mb=64)
    ctx = create_coll_context(max_seg_elems=1 << 16, max_ll_words=4096)

    g = torch.Generator().manual_seed(11)
    xs = [torch.randn(world * 128, 64, generator=g).to(torch.bfloat16)
          for _ in range(world)]
    x = xs[rank].cuda()
    for it in range(3):  # repeat: monotonic-tag protocol needs no resets
        out = reduce_scatter(x, ctx)
        torch.cuda.synchronize()
        ref = sum(t.float() for t in xs).reshape(world, 128, 64)[rank]
        err = (out.float().cpu() - ref).abs().max().item()
        assert err < 0.5, (it, err)

    y = (torch.arange(512, dtype=torch.float32).reshape(4, 128)
         + rank * 1000).cuda()
    for it in range(3):
        gathered = ll_all_gather(y, ctx)
        torch.cuda.synchronize()
        got = gathered.cpu()
        for r in range(world):
            exp = (torch.arange(512, dtype=torch.float32).reshape(4, 128)
                   + r * 1000)
            assert torch.equal(got[r * 4:(r + 1) * 4], exp), (it, r)
    z = ((torch.arange(world * 64 * 16, dtype=torch.float32) % 101)
         .reshape(world * 64, 16).to(torch.bfloat16) + rank * 7).cuda()
    for it in range(2):
        a2a = all_to_all_single(z, ctx)
        torch.cuda.synchronize()
        got = a2a.float().cpu().reshape(world, 64, 16)
        zc = z.float().cpu().reshape(world, 64, 16)
        for p in range(world):
            exp = zc[rank] + (p - rank) * 7
            assert torch.equal(got[p], exp), (it, rank, p)
    td.shutdown_heap()


def test_collectives_2rank():
    run_distributed(_body_collectives, world_size=2)
