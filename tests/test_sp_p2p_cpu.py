"""CPU tests: SP flash-decode LSE merge, Ulysses a2a, p2p ring (gloo+shm)."""
import torch

from tests.conftest import run_distributed


def _body_sp_decode(rank, world):
    from triton_dist_amd.ops import (create_sp_flash_decode_context,
                                     sp_flash_decode, sp_flash_decode_ref)
    from triton_dist_amd.utils import assert_allclose

    b, qh, kvh, d = 4, 4, 2, 128
    chunk, maxlen = 16, 32
    ctx = create_sp_flash_decode_context(max_batch=b, qh=qh)
    g = torch.Generator().manual_seed(3)  # same full KV on all ranks
    k_full = (torch.randn(b, world * chunk, kvh, d, generator=g) / 4).to(torch.bfloat16)
    v_full = (torch.randn(b, world * chunk, kvh, d, generator=g) / 4).to(torch.bfloat16)
    q = (torch.randn(b, qh * d, generator=g) / 4).to(torch.bfloat16)
    # my shard
    kc = torch.zeros(b, maxlen, kvh, d, dtype=torch.bfloat16)
    vc = torch.zeros(b, maxlen, kvh, d, dtype=torch.bfloat16)
    kc[:, :chunk] = k_full[:, rank * chunk:(rank + 1) * chunk]
    vc[:, :chunk] = v_full[:, rank * chunk:(rank + 1) * chunk]
    clen = torch.tensor(chunk, dtype=torch.int64)
    out = sp_flash_decode(q, kc, vc, clen, ctx, qh, kvh)
    ref = sp_flash_decode_ref(q, k_full, v_full, world * chunk, qh, kvh)
    assert_allclose(out, ref, atol=5e-2, rtol=5e-2)


def test_sp_flash_decode_cpu_2rank():
    run_distributed(_body_sp_decode, world_size=2)


def _body_ulysses(rank, world):
    from triton_dist_amd.ops import (create_ulysses_context, ulysses_a2a,
                                     ulysses_a2a_ref)
    from triton_dist_amd.utils import assert_allclose

    t_loc, heads, d = 8, 4 * world, 16
    ctx = create_ulysses_context(max_tokens=t_loc, n_heads=heads, head_dim=d)
    g = torch.Generator().manual_seed(10 + rank)
    x = (torch.randn(t_loc, heads, d, generator=g)).to(torch.bfloat16)
    out = ulysses_a2a(x, ctx)
    ref = ulysses_a2a_ref(x)
    assert torch.equal(out, ref), (out.shape, ref.shape)


def test_ulysses_cpu_2rank():
    run_distributed(_body_ulysses, world_size=2)


def test_ulysses_cpu_4rank():
    run_distributed(_body_ulysses, world_size=4)


def _body_p2p(rank, world):
    from triton_dist_amd.ops import create_p2p_context, p2p_recv, p2p_send

    ctx = create_p2p_context(max_bytes=4096, depth=2)
    nxt, prv = (rank + 1) % world, (rank - 1) % world
    for i in range(6):  # exceed depth to exercise credits
        x = torch.full((64,), float(rank * 10 + i)).to(torch.bfloat16)
        p2p_send(x, nxt, ctx)
        out = torch.empty(64, dtype=torch.bfloat16)
        p2p_recv(out, prv, ctx)
        assert (out.float() == prv * 10 + i).all(), (rank, i, out[0])


def test_p2p_ring_cpu_2rank():
    run_distributed(_body_p2p, world_size=2)


def test_p2p_ring_cpu_4rank():
    run_distributed(_body_p2p, world_size=4)


def _body_ulysses_roundtrip(rank, world):
    from triton_dist_amd.layers import UlyssesSPAllToAllLayer

    t_loc, heads, d = 6, 4 * world, 16
    layer = UlyssesSPAllToAllLayer(heads, d)
    layer.init_ctx(t_loc)
    g = torch.Generator().manual_seed(4 + rank)
    x = torch.randn(t_loc, heads, d, generator=g).to(torch.bfloat16)
    y = layer.pre_attn(x)            # [world*t_loc, heads/world, d]
    assert y.shape == (world * t_loc, heads // world, d)
    back = layer.post_attn(y.clone())
    assert torch.equal(back, x), "ulysses pre+post must round-trip"


def test_ulysses_layer_roundtrip_cpu_2rank():
    run_distributed(_body_ulysses_roundtrip, world_size=2)


def test_sp_decode_cpu_4rank():
    run_distributed(_body_sp_decode, world_size=4)


def _body_ulysses_fused(rank, world):
    import torch
    from triton_dist_amd.ops import (create_ulysses_fused_context,
                                     ulysses_a2a_o_gemm,
                                     ulysses_qkv_gemm_a2a)
    from triton_dist_amd.utils import assert_allclose, rand_tensor

    t_loc, hdim = 8, 64
    qkv_dim = world * 512   # peer_cols = 512 (tiles by 256)
    o_in = 512
    n_out = 256
    ctx = create_ulysses_fused_context(t_loc, qkv_dim, o_in)
    g = torch.Generator().manual_seed(5 + rank)
    x = rand_tensor((t_loc, hdim), dtype=torch.bfloat16, generator=g) / 8
    gw = torch.Generator().manual_seed(42)
    w_qkv = rand_tensor((qkv_dim, hdim), dtype=torch.bfloat16,
                        generator=gw) / 8
    w_o_split = rand_tensor((world, n_out, o_in), dtype=torch.bfloat16,
                            generator=gw) / 8
    import torch.distributed as dist
    for _ in range(2):
        mine = ulysses_qkv_gemm_a2a(x, w_qkv, ctx)
        # golden: gather everyone's x, project, slice my column block
        xs = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(xs, x)
        full = torch.cat(xs, 0).float() @ w_qkv.float().t()
        pc = qkv_dim // world
        ref = full[:, rank * pc:(rank + 1) * pc]
        assert_allclose(mine, ref.to(torch.bfloat16), atol=8e-2, rtol=5e-2)

        # o side: use `mine` (shape [world*t_loc, pc]) truncated to o_in
        attn = mine[:, :o_in].contiguous()
        out = ulysses_a2a_o_gemm(attn, w_o_split, ctx)
        # golden: each rank's tokens gather their head shards from all
        # ranks; o = sum_src attn_src(tokens of mine) @ w_o_split[src]^T
        att_all = [torch.empty_like(attn) for _ in range(world)]
        dist.all_gather(att_all, attn)
        acc = torch.zeros(t_loc, n_out, dtype=torch.float32)
        for src in range(world):
            seg = att_all[src][rank * t_loc:(rank + 1) * t_loc].float()
            acc += seg @ w_o_split[src].float().t()
        assert_allclose(out, acc.to(torch.bfloat16), atol=8e-2, rtol=5e-2)


def test_ulysses_fused_cpu_2rank():
    run_distributed(_body_ulysses_fused, world_size=2)


def test_ulysses_fused_cpu_4rank():
    run_distributed(_body_ulysses_fused, world_size=4)


def _body_zigzag(rank, world):
    import torch
    import torch.nn.functional as F
    import torch.distributed as dist
    from triton_dist_amd.ops import (create_sp_ag_attn_context,
                                     sp_ag_attention_zigzag)
    from triton_dist_amd.utils import assert_allclose

    s_blk, qh, kvh, d = 8, 4, 2, 128
    nb = 2 * world
    ctx = create_sp_ag_attn_context(2 * s_blk, kvh, d)
    torch.manual_seed(7)  # same full sequence everywhere
    kf = (torch.randn(nb * s_blk, kvh, d) / 4).to(torch.bfloat16)
    vf = (torch.randn(nb * s_blk, kvh, d) / 4).to(torch.bfloat16)
    qf = (torch.randn(nb * s_blk, qh, d) / 4).to(torch.bfloat16)

    def my_blocks(x):
        b = x.reshape(nb, s_blk, x.shape[1], d)
        return torch.stack([b[rank], b[nb - 1 - rank]], 0)

    out = sp_ag_attention_zigzag(my_blocks(qf), my_blocks(kf),
                                 my_blocks(vf), ctx, qh)
    # golden: full causal attention, sliced at my two blocks
    qt = qf.permute(1, 0, 2).unsqueeze(0).float()
    kt = kf.permute(1, 0, 2).unsqueeze(0).float()
    vt = vf.permute(1, 0, 2).unsqueeze(0).float()
    ref = F.scaled_dot_product_attention(qt, kt, vt, is_causal=True,
                                         enable_gqa=True)
    ref = ref.squeeze(0).permute(1, 0, 2).reshape(nb, s_blk, qh, d)
    want = torch.stack([ref[rank], ref[nb - 1 - rank]], 0)
    assert_allclose(out, want.to(torch.bfloat16), atol=8e-2, rtol=5e-2)


def test_sp_zigzag_cpu_2rank():
    run_distributed(_body_zigzag, world_size=2)


def test_sp_zigzag_cpu_4rank():
    run_distributed(_body_zigzag, world_size=4)
