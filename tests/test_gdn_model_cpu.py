"""Hybrid GDN+attention model (Qwen3-Next geometry): dist ag_rs forward
vs the replicated torch golden, prefill + decode, 1/2 ranks; Engine
serve smoke with the hybrid cache."""
import torch

from tests.conftest import run_distributed


def _body_hybrid(rank, world, model_name="tiny-gdn"):
    import triton_dist_amd as td
    from triton_dist_amd.layers.gdn_layer import GDNMixer
    from triton_dist_amd.models import AutoLLM, HybridGDNLLM, get_config
    from triton_dist_amd.utils import assert_allclose

    td.init_symm_heap(size_mb=32)
    cfg = get_config(model_name, tp_mode="ag_rs", max_length=64)
    model = AutoLLM(cfg, device="cpu")
    assert isinstance(model, HybridGDNLLM)
    kinds = [type(l["attn"]).__name__ for l in model.layers]
    assert kinds == ["GDNMixer", "GDNMixer", "TP_Attn"], kinds
    model.init_weights(seed=5)
    b, s = world * 2, 8
    model.init_dist_ctx(max_m_total=b * s)

    kv1 = model.make_cache(b, 32)
    kv2 = model.make_cache(b, 32)
    tokens = torch.randint(0, cfg.vocab, (b, s),
                           generator=torch.Generator().manual_seed(2))
    pos = torch.arange(s).expand(b, s)
    logits1 = model.step(tokens, kv1, pos, prefill=True)
    logits2 = model.step(tokens, kv2, pos, prefill=True, mode="torch")
    assert_allclose(logits1, logits2, atol=1e-1, rtol=5e-2)
    kv1.advance(s)
    kv2.advance(s)

    nxt = logits1.argmax(-1).view(b, 1)
    p1 = kv1.offset.reshape(1, 1).expand(b, 1)
    d1 = model.step(nxt, kv1, p1, prefill=False)
    d2 = model.step(nxt, kv2, p1, prefill=False, mode="torch")
    assert_allclose(d1, d2, atol=1e-1, rtol=5e-2)
    td.shutdown_heap()


def test_hybrid_gdn_model_1rank():
    run_distributed(_body_hybrid, world_size=1)


def test_hybrid_gdn_model_2rank():
    run_distributed(_body_hybrid, world_size=2)


def _body_engine(rank, world):
    import triton_dist_amd as td
    from triton_dist_amd.models import AutoLLM, Engine, get_config
    from triton_dist_amd.models.kv_cache import HybridCache

    td.init_symm_heap(size_mb=32)
    cfg = get_config("tiny-gdn", tp_mode="ag_rs", max_length=64)
    model = AutoLLM(cfg, device="cpu")
    model.init_weights(seed=5)
    model.init_dist_ctx(max_m_total=world * 2 * 8)
    eng = Engine(model, batch=world * 2, max_len=32, use_graph=False)
    assert isinstance(eng.kv, HybridCache)
    prompt = torch.randint(0, cfg.vocab, (world * 2, 8))
    out = eng.serve(prompt, gen_len=4)
    assert out.shape == (world * 2, 4)
    # GDN state advanced (non-zero) for a mixer layer
    st = eng.kv.gdn_state(0, world * 2, model.layers[0]["attn"].lh,
                          cfg.gdn_head_k, cfg.gdn_head_v)
    assert st.abs().sum() > 0
    td.shutdown_heap()


def test_hybrid_gdn_engine_serve():
    run_distributed(_body_engine, world_size=2)


def _body_hybrid4(rank, world):
    _body_hybrid(rank, world, model_name="tiny-gdn4")


def test_hybrid_gdn_model_4rank():
    run_distributed(_body_hybrid4, world_size=4)


def _body_hybrid_armode(rank, world):
    import triton_dist_amd as td
    from triton_dist_amd.models import AutoLLM, get_config
    from triton_dist_amd.utils import assert_allclose

    td.init_symm_heap(size_mb=32)
    cfg = get_config("tiny-gdn", tp_mode="allreduce", max_length=64)
    model = AutoLLM(cfg, device="cpu")
    model.init_weights(seed=5)
    b, s = 4, 8  # replicated modes: full batch on every rank
    model.init_dist_ctx(max_m_total=b * s)
    kv1 = model.make_cache(b, 32)
    kv2 = model.make_cache(b, 32)
    tokens = torch.randint(0, cfg.vocab, (b, s),
                           generator=torch.Generator().manual_seed(3))
    pos = torch.arange(s).expand(b, s)
    l1 = model.step(tokens, kv1, pos, prefill=True)
    l2 = model.step(tokens, kv2, pos, prefill=True, mode="torch")
    assert_allclose(l1, l2, atol=1e-1, rtol=5e-2)


def test_hybrid_gdn_allreduce_mode_2rank():
    run_distributed(_body_hybrid_armode, world_size=2)


def _body_hybrid_gemm_ar(rank, world):
    import triton_dist_amd as td
    from triton_dist_amd.models import AutoLLM, get_config
    from triton_dist_amd.utils import assert_allclose

    td.init_symm_heap(size_mb=32)
    cfg = get_config("tiny-gdn", tp_mode="gemm_ar", max_length=64)
    model = AutoLLM(cfg, device="cpu")
    model.init_weights(seed=5)
    b, s = 4, 8
    model.init_dist_ctx(max_m_total=b * s)
    kv1 = model.make_cache(b, 32)
    kv2 = model.make_cache(b, 32)
    tokens = torch.randint(0, cfg.vocab, (b, s),
                           generator=torch.Generator().manual_seed(3))
    pos = torch.arange(s).expand(b, s)
    l1 = model.step(tokens, kv1, pos, prefill=True)
    l2 = model.step(tokens, kv2, pos, prefill=True, mode="torch")
    assert_allclose(l1, l2, atol=1e-1, rtol=5e-2)


def test_hybrid_gdn_gemm_ar_mode_2rank():
    run_distributed(_body_hybrid_gemm_ar, world_size=2)
