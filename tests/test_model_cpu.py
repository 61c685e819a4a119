"""CPU model tests: tiny-config DenseLLM, distributed ag_rs decode vs the
torch-eager golden path (the reference repo's test_tp_e2e.py --check
pattern) on gloo."""
import torch

from tests.conftest import run_distributed


def _body_model_modes(rank, world, model_name="tiny"):
    from triton_dist_amd.models import DenseLLM, Engine, KVCache, get_config
    from triton_dist_amd.utils import assert_allclose

    cfg = get_config(model_name, tp_mode="ag_rs", max_length=64)
    model = DenseLLM(cfg, device="cpu")
    model.init_weights(seed=3)
    b, s = world * 2, 4
    model.init_dist_ctx(max_m_total=b)

    kv1 = KVCache(cfg.n_layers, b, 32, cfg.n_kv_heads // world, cfg.head_dim)
    kv2 = KVCache(cfg.n_layers, b, 32, cfg.n_kv_heads // world, cfg.head_dim)
    tokens = torch.randint(0, cfg.vocab, (b, s),
                           generator=torch.Generator().manual_seed(1))

    # prefill identically on both paths
    first1 = model.prefill(tokens, kv1)
    first2 = model.prefill(tokens, kv2)
    assert torch.equal(first1, first2)

    # decode: dist ag_rs path vs torch path, logits must agree
    pos = kv1.offset.reshape(1, 1).expand(b, 1)
    logits_dist = model.step(first1.view(b, 1), kv1, pos, prefill=False)
    pos2 = kv2.offset.reshape(1, 1).expand(b, 1)
    logits_ref = model.step(first2.view(b, 1), kv2, pos2, prefill=False,
                            mode="torch")
    assert_allclose(logits_dist, logits_ref, atol=1e-1, rtol=5e-2)


def test_model_dist_vs_torch_2rank():
    run_distributed(_body_model_modes, world_size=2)


def _body_engine_serve(rank, world):
    from triton_dist_amd.models import DenseLLM, Engine, get_config

    cfg = get_config("tiny", tp_mode="ag_rs", max_length=64)
    model = DenseLLM(cfg, device="cpu")
    model.init_weights(seed=4)
    b, s, gen = world * 2, 4, 5
    model.init_dist_ctx(max_m_total=b)
    eng = Engine(model, batch=b, max_len=32, use_graph=False)
    out = eng.serve(torch.randint(0, cfg.vocab, (b, s),
                                  generator=torch.Generator().manual_seed(2)),
                    gen_len=gen)
    assert out.shape == (b, gen)
    # all ranks must produce identical tokens (replicated sampling)
    import torch.distributed as dist

    gathered = [torch.empty_like(out) for _ in range(world)]
    dist.all_gather(gathered, out)
    for g in gathered:
        assert torch.equal(g, out), "ranks diverged"


def test_engine_serve_2rank():
    run_distributed(_body_engine_serve, world_size=2)


def _body_model_modes4(rank, world):
    _body_model_modes(rank, world, model_name="tiny4")


def test_model_modes_cpu_4rank():
    run_distributed(_body_model_modes4, world_size=4)


def test_paged_kv_cache():
    """Paged pool semantics: append across block boundaries, gather
    equals a dense reference cache."""
    import torch

    from triton_dist_amd.models.kv_cache import KVCache, PagedKVCache

    L, B, MAX, KVH, D, BLK = 2, 3, 128, 2, 8, 32
    dense = KVCache(L, B, MAX, KVH, D)
    paged = PagedKVCache(L, B, MAX, KVH, D, block=BLK)
    g = torch.Generator().manual_seed(0)
    pos = 0
    for step_len in (40, 1, 1, 25):  # crosses block boundaries
        for layer in range(L):
            k = torch.randn(B, step_len, KVH, D, generator=g).to(
                torch.bfloat16)
            v = torch.randn(B, step_len, KVH, D, generator=g).to(
                torch.bfloat16)
            dense.k[layer, :, pos:pos + step_len].copy_(k)
            dense.v[layer, :, pos:pos + step_len].copy_(v)
            paged.append(layer, k, v, pos)
        pos += step_len
    for layer in range(L):
        pk, pv = paged.gather_layer(layer, pos)
        assert torch.equal(pk, dense.k[layer, :, :pos])
        assert torch.equal(pv, dense.v[layer, :, :pos])
    # pool economy: only the blocks actually touched were allocated
    assert paged._free_top == B * ((pos + BLK - 1) // BLK)
    paged.reset()
    assert (paged.block_table < 0).all()


def test_flash_decode_paged_cpu_bridge():
    """The paged op's CPU path (gather bridge + torch flash reference)
    agrees with the contiguous op."""
    import torch

    from triton_dist_amd.models.kv_cache import PagedKVCache
    from triton_dist_amd.ops.fused import flash_decode_paged_op

    torch.manual_seed(4)
    b, qh, kvh, d, maxlen, seq = 2, 4, 2, 32, 128, 37
    paged = PagedKVCache(1, b, maxlen, kvh, d, block=32)
    kc = (torch.randn(b, maxlen, kvh, d) / 4).to(torch.bfloat16)
    vc = (torch.randn(b, maxlen, kvh, d) / 4).to(torch.bfloat16)
    paged.append(0, kc[:, :seq], vc[:, :seq], 0)
    q = (torch.randn(b, qh * d) / 4).to(torch.bfloat16)
    offset = torch.tensor(seq - 1, dtype=torch.int64)
    out_p = flash_decode_paged_op(q, paged, 0, offset, qh, kvh)
    import torch.nn.functional as F
    qs = q.view(b, qh, 1, d).float()
    ks = kc[:, :seq].transpose(1, 2).float()
    vs = vc[:, :seq].transpose(1, 2).float()
    ref = F.scaled_dot_product_attention(qs, ks, vs, enable_gqa=True)
    assert (out_p.float() - ref.view(b, qh * d)).abs().max() < 2e-2
