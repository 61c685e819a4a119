"""GPU SP + P2P tests (2 ranks sharing one GPU over hipIpc)."""
import pytest
import torch

from tests.conftest import run_distributed

pytestmark = pytest.mark.gpu


def _body_sp_decode(rank, world):
    from triton_dist_amd.ops import (create_sp_flash_decode_context,
                                     sp_flash_decode, sp_flash_decode_ref)
    from triton_dist_amd.utils import assert_allclose

    b, qh, kvh, d = 16, 8, 2, 128
    chunk, maxlen = 48, 64
    ctx = create_sp_flash_decode_context(max_batch=b, qh=qh)
    g = torch.Generator("cuda").manual_seed(3)
    k_full = (torch.randn(b, world * chunk, kvh, d, device="cuda",
                          generator=g) / 4).to(torch.bfloat16)
    v_full = (torch.randn(b, world * chunk, kvh, d, device="cuda",
                          generator=g) / 4).to(torch.bfloat16)
    q = (torch.randn(b, qh * d, device="cuda", generator=g) / 4
         ).to(torch.bfloat16)
    kc = torch.zeros(b, maxlen, kvh, d, dtype=torch.bfloat16, device="cuda")
    vc = torch.zeros(b, maxlen, kvh, d, dtype=torch.bfloat16, device="cuda")
    kc[:, :chunk] = k_full[:, rank * chunk:(rank + 1) * chunk]
    vc[:, :chunk] = v_full[:, rank * chunk:(rank + 1) * chunk]
    clen = torch.tensor(chunk, dtype=torch.int64, device="cuda")
    for _ in range(2):
        out = sp_flash_decode(q, kc, vc, clen, ctx, qh, kvh)
        torch.cuda.synchronize()
        ref = sp_flash_decode_ref(q, k_full, v_full, world * chunk, qh, kvh)
        assert_allclose(out, ref, atol=5e-2, rtol=5e-2)


def test_sp_flash_decode_gpu_2rank():
    run_distributed(_body_sp_decode, world_size=2)


def _body_ulysses(rank, world):
    from triton_dist_amd.ops import (create_ulysses_context, ulysses_a2a,
                                     ulysses_a2a_ref)

    t_loc, heads, d = 64, 8, 128
    ctx = create_ulysses_context(max_tokens=t_loc, n_heads=heads, head_dim=d)
    torch.manual_seed(10 + rank)
    x = torch.randn(t_loc, heads, d, device="cuda").to(torch.bfloat16)
    for _ in range(2):
        out = ulysses_a2a(x, ctx)
        torch.cuda.synchronize()
        ref = ulysses_a2a_ref(x)
        assert torch.equal(out.cpu(), ref.cpu())


def test_ulysses_gpu_2rank():
    run_distributed(_body_ulysses, world_size=2)


def _body_ag_attn(rank, world):
    import torch.nn.functional as F

    from triton_dist_amd.ops import create_sp_ag_attn_context, sp_ag_attention
    from triton_dist_amd.utils import assert_allclose

    s_loc, qh, kvh, d = 128, 8, 2, 128
    ctx = create_sp_ag_attn_context(max_chunk_tokens=s_loc, kvh=kvh,
                                    head_dim=d)
    g = torch.Generator("cuda").manual_seed(1)  # same full tensors everywhere
    total = world * s_loc
    q_full = (torch.randn(total, qh, d, device="cuda", generator=g) / 4
              ).to(torch.bfloat16)
    k_full = (torch.randn(total, kvh, d, device="cuda", generator=g) / 4
              ).to(torch.bfloat16)
    v_full = (torch.randn(total, kvh, d, device="cuda", generator=g) / 4
              ).to(torch.bfloat16)
    q = q_full[rank * s_loc:(rank + 1) * s_loc]
    out = sp_ag_attention(q, k_full[rank * s_loc:(rank + 1) * s_loc],
                          v_full[rank * s_loc:(rank + 1) * s_loc], ctx, qh)
    torch.cuda.synchronize()
    # golden: full causal attention, slice my rows
    ref = F.scaled_dot_product_attention(
        q_full.permute(1, 0, 2).unsqueeze(0).float(),
        k_full.permute(1, 0, 2).unsqueeze(0).float(),
        v_full.permute(1, 0, 2).unsqueeze(0).float(),
        is_causal=True, enable_gqa=True)
    ref = ref.squeeze(0).permute(1, 0, 2)[rank * s_loc:(rank + 1) * s_loc]
    assert_allclose(out, ref, atol=5e-2, rtol=5e-2)


def test_sp_ag_attention_gpu_2rank():
    run_distributed(_body_ag_attn, world_size=2)


def _body_p2p(rank, world):
    from triton_dist_amd.ops import create_p2p_context, p2p_recv, p2p_send

    ctx = create_p2p_context(max_bytes=1 << 20, depth=2)
    nxt, prv = (rank + 1) % world, (rank - 1) % world
    for i in range(6):
        x = torch.full((4096,), float(rank * 10 + i), device="cuda"
                       ).to(torch.bfloat16)
        p2p_send(x, nxt, ctx)
        out = torch.empty(4096, dtype=torch.bfloat16, device="cuda")
        p2p_recv(out, prv, ctx)
        torch.cuda.synchronize()
        assert (out.float() == prv * 10 + i).all(), (rank, i)


def test_p2p_ring_gpu_2rank():
    run_distributed(_body_p2p, world_size=2)


def _body_ulysses_fused_gpu(rank, world):
    import torch.distributed as dist
    from triton_dist_amd.ops import (create_ulysses_fused_context,
                                     ulysses_a2a_o_gemm,
                                     ulysses_qkv_gemm_a2a)
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    t_loc, hdim = 256, 1024
    qkv_dim = world * 1280   # qwen3-32b-like per-rank block (5 x 256)
    o_in = 1024
    n_out = 1024
    ctx = create_ulysses_fused_context(t_loc, qkv_dim, o_in)
    torch.manual_seed(3 + rank)
    x = (torch.randn(t_loc, hdim, device="cuda") / 8).to(torch.bfloat16)
    torch.manual_seed(77)
    w_qkv = (torch.randn(qkv_dim, hdim, device="cuda") / 8
             ).to(torch.bfloat16)
    w_o_split = (torch.randn(world, n_out, o_in, device="cuda") / 8
                 ).to(torch.bfloat16)
    for _ in range(2):
        mine = ulysses_qkv_gemm_a2a(x, w_qkv, ctx)
        torch.cuda.synchronize()
        xs = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(xs, x)
        full = torch.cat(xs, 0).float() @ w_qkv.float().t()
        pc = qkv_dim // world
        ref = full[:, rank * pc:(rank + 1) * pc]
        assert_allclose(mine, ref.to(torch.bfloat16), **bf16_gemm_tol(hdim))

        attn = mine[:, :o_in].contiguous()
        out = ulysses_a2a_o_gemm(attn, w_o_split, ctx)
        torch.cuda.synchronize()
        att_all = [torch.empty_like(attn) for _ in range(world)]
        dist.all_gather(att_all, attn)
        acc = torch.zeros(t_loc, n_out, dtype=torch.float32, device="cuda")
        for src in range(world):
            seg = att_all[src][rank * t_loc:(rank + 1) * t_loc].float()
            acc += seg @ w_o_split[src].float().t()
        assert_allclose(out, acc.to(torch.bfloat16), atol=2.5e-1, rtol=5e-2)


def test_ulysses_fused_gpu_2rank():
    run_distributed(_body_ulysses_fused_gpu, world_size=2)


def _body_zigzag_gpu(rank, world):
    import torch.nn.functional as F
    from triton_dist_amd.ops import (create_sp_ag_attn_context,
                                     sp_ag_attention_zigzag)
    from triton_dist_amd.utils import assert_allclose

    s_blk, qh, kvh, d = 64, 8, 2, 128
    nb = 2 * world
    ctx = create_sp_ag_attn_context(2 * s_blk, kvh, d)
    torch.manual_seed(7)
    kf = (torch.randn(nb * s_blk, kvh, d, device="cuda") / 4
          ).to(torch.bfloat16)
    vf = (torch.randn(nb * s_blk, kvh, d, device="cuda") / 4
          ).to(torch.bfloat16)
    qf = (torch.randn(nb * s_blk, qh, d, device="cuda") / 4
          ).to(torch.bfloat16)

    def my_blocks(x):
        b = x.reshape(nb, s_blk, x.shape[1], d)
        return torch.stack([b[rank], b[nb - 1 - rank]], 0)

    for _ in range(2):
        out = sp_ag_attention_zigzag(my_blocks(qf), my_blocks(kf),
                                     my_blocks(vf), ctx, qh)
        torch.cuda.synchronize()
        qt = qf.permute(1, 0, 2).unsqueeze(0).float()
        kt = kf.permute(1, 0, 2).unsqueeze(0).float()
        vt = vf.permute(1, 0, 2).unsqueeze(0).float()
        ref = F.scaled_dot_product_attention(qt, kt, vt, is_causal=True,
                                             enable_gqa=True)
        ref = ref.squeeze(0).permute(1, 0, 2).reshape(nb, s_blk, qh, d)
        want = torch.stack([ref[rank], ref[nb - 1 - rank]], 0)
        assert_allclose(out, want.to(torch.bfloat16), atol=8e-2, rtol=5e-2)


def test_sp_zigzag_gpu_2rank():
    run_distributed(_body_zigzag_gpu, world_size=2)
