"""GPU EP MoE tests: 2 ranks sharing a GPU, full dispatch/grouped-GEMM/
combine pipeline vs golden reference."""
import pytest
import torch

from tests.conftest import run_distributed

pytestmark = pytest.mark.gpu


def _body_ep_gpu(rank, world):
    from triton_dist_amd.ops import create_ep_context, ep_moe_forward, ep_moe_ref
    from triton_dist_amd.utils import assert_allclose

    T, H, inter, E, K = 64, 512, 128, 2 * world, 2
    e_loc = E // world
    ctx = create_ep_context(max_tokens=T, hidden=H, n_experts=E, topk=K)
    g = torch.Generator("cuda").manual_seed(7)
    full_gu = (torch.randn(E, 2 * inter, H, device="cuda", generator=g)
               * 0.1).to(torch.bfloat16)
    full_d = (torch.randn(E, H, inter, device="cuda", generator=g)
              * 0.1).to(torch.bfloat16)
    w_gu = full_gu[rank * e_loc:(rank + 1) * e_loc].contiguous()
    w_d = full_d[rank * e_loc:(rank + 1) * e_loc].contiguous()

    gt = torch.Generator("cuda").manual_seed(100 + rank)
    x = (torch.randn(T, H, device="cuda", generator=gt) / 4).to(torch.bfloat16)
    logits = torch.randn(T, E, device="cuda", generator=gt)
    topk_w32, topk_ids = torch.topk(torch.softmax(logits, -1), K, dim=-1)
    topk_ids = topk_ids.to(torch.int32).contiguous()
    topk_w = topk_w32.float().contiguous()

    for it in range(3):
        out = ep_moe_forward(x, topk_ids, topk_w, w_gu, w_d, ctx)
        torch.cuda.synchronize()
        ref = ep_moe_ref(x, topk_ids, topk_w, full_gu, full_d)
        assert_allclose(out, ref, atol=8e-2, rtol=8e-2, msg=f"iter {it}")


def test_ep_moe_gpu_2rank():
    run_distributed(_body_ep_gpu, world_size=2)


def _body_ep_ll_gpu(rank, world):
    from triton_dist_amd.ops import create_ep_context, ep_moe_forward, ep_moe_ref
    from triton_dist_amd.utils import assert_allclose

    T, H, inter, E, K = 64, 512, 128, 2 * world, 2
    e_loc = E // world
    ctx = create_ep_context(max_tokens=T, hidden=H, n_experts=E, topk=K,
                            low_latency=True)
    g = torch.Generator("cuda").manual_seed(7)
    full_gu = (torch.randn(E, 2 * inter, H, device="cuda", generator=g)
               * 0.1).to(torch.bfloat16)
    full_d = (torch.randn(E, H, inter, device="cuda", generator=g)
              * 0.1).to(torch.bfloat16)
    w_gu = full_gu[rank * e_loc:(rank + 1) * e_loc].contiguous()
    w_d = full_d[rank * e_loc:(rank + 1) * e_loc].contiguous()
    gt = torch.Generator("cuda").manual_seed(100 + rank)
    x = (torch.randn(T, H, device="cuda", generator=gt) / 4).to(torch.bfloat16)
    logits = torch.randn(T, E, device="cuda", generator=gt)
    topk_w32, topk_ids = torch.topk(torch.softmax(logits, -1), K, dim=-1)
    topk_ids = topk_ids.to(torch.int32).contiguous()
    topk_w = topk_w32.float().contiguous()
    ref = ep_moe_ref(x, topk_ids, topk_w, full_gu, full_d)
    for it in range(5):  # parity cycling + credit reuse
        out = ep_moe_forward(x, topk_ids, topk_w, w_gu, w_d, ctx)
        torch.cuda.synchronize()
        assert_allclose(out, ref, atol=8e-2, rtol=8e-2, msg=f"ll iter {it}")


def test_ep_moe_ll_gpu_2rank():
    run_distributed(_body_ep_ll_gpu, world_size=2)


def _body_ep_fp8_gpu(rank, world):
    from triton_dist_amd.ops import create_ep_context, ep_moe_forward, ep_moe_ref
    from triton_dist_amd.utils import assert_allclose

    T, H, inter, E, K = 64, 512, 128, 2 * world, 2
    e_loc = E // world
    ctx = create_ep_context(max_tokens=T, hidden=H, n_experts=E, topk=K,
                            low_latency=True, fp8=True)
    g = torch.Generator("cuda").manual_seed(7)
    full_gu = (torch.randn(E, 2 * inter, H, device="cuda", generator=g)
               * 0.1).to(torch.bfloat16)
    full_d = (torch.randn(E, H, inter, device="cuda", generator=g)
              * 0.1).to(torch.bfloat16)
    w_gu = full_gu[rank * e_loc:(rank + 1) * e_loc].contiguous()
    w_d = full_d[rank * e_loc:(rank + 1) * e_loc].contiguous()
    gt = torch.Generator("cuda").manual_seed(100 + rank)
    x = (torch.randn(T, H, device="cuda", generator=gt) / 4).to(torch.bfloat16)
    logits = torch.randn(T, E, device="cuda", generator=gt)
    topk_w32, topk_ids = torch.topk(torch.softmax(logits, -1), K, dim=-1)
    topk_ids = topk_ids.to(torch.int32).contiguous()
    topk_w = topk_w32.float().contiguous()
    ref = ep_moe_ref(x, topk_ids, topk_w, full_gu, full_d)
    for it in range(3):
        out = ep_moe_forward(x, topk_ids, topk_w, w_gu, w_d, ctx)
        torch.cuda.synchronize()
        # fp8 payload: wider tolerance (e4m3 ~2 decimal digits)
        assert_allclose(out, ref, atol=1.5e-1, rtol=1.5e-1,
                        msg=f"fp8 iter {it}")


def test_ep_moe_fp8_gpu_2rank():
    run_distributed(_body_ep_fp8_gpu, world_size=2)
