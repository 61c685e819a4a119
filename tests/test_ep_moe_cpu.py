"""CPU tests of the EP MoE dispatch/grouped-FFN/combine pipeline (gloo +
shm mock) vs the single-device golden reference."""
import torch

from tests.conftest import run_distributed


def _body_ep(rank, world):
    from triton_dist_amd.ops import create_ep_context, ep_moe_forward, ep_moe_ref
    from triton_dist_amd.utils import assert_allclose

    T, H, inter, E, K = 8, 32, 16, 2 * world, 2
    e_loc = E // world
    ctx = create_ep_context(max_tokens=T, hidden=H, n_experts=E, topk=K)
    g = torch.Generator().manual_seed(7)
    full_gu = (torch.randn(E, 2 * inter, H, generator=g) * 0.1).to(torch.bfloat16)
    full_d = (torch.randn(E, H, inter, generator=g) * 0.1).to(torch.bfloat16)
    w_gu = full_gu[rank * e_loc:(rank + 1) * e_loc].contiguous()
    w_d = full_d[rank * e_loc:(rank + 1) * e_loc].contiguous()

    gt = torch.Generator().manual_seed(100 + rank)
    x = (torch.randn(T, H, generator=gt) / 4).to(torch.bfloat16)
    logits = torch.randn(T, E, generator=gt)
    topk_w32, topk_ids = torch.topk(torch.softmax(logits, -1), K, dim=-1)
    topk_ids = topk_ids.to(torch.int32)
    topk_w = topk_w32.float().contiguous()

    for _ in range(2):
        out = ep_moe_forward(x, topk_ids, topk_w, w_gu, w_d, ctx)
        ref = ep_moe_ref(x, topk_ids, topk_w, full_gu, full_d)
        assert_allclose(out, ref, atol=8e-2, rtol=8e-2)


def test_ep_moe_cpu_2rank():
    run_distributed(_body_ep, world_size=2)


def test_ep_moe_cpu_4rank():
    run_distributed(_body_ep, world_size=4)
