"""Megakernel decode step vs the eager layer path (single GPU)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_megakernel_decode_matches_eager():
    import triton_dist_amd as td
    from triton_dist_amd.mega import MegaQwen3Decode
    from triton_dist_amd.models import DenseLLM, KVCache, get_config
    from triton_dist_amd.utils import assert_allclose

    td.initialize_distributed()
    td.init_symm_heap()
    cfg = get_config("tiny-gpu", tp_mode="ag_rs", max_length=128)
    model = DenseLLM(cfg, device="cuda")
    model.init_weights(seed=5)
    b = 64
    model.init_dist_ctx(max_m_total=128)  # for the eager reference path

    kv_ref = KVCache(cfg.n_layers, b, 64, cfg.n_kv_heads, cfg.head_dim,
                     device="cuda")
    kv_meg = KVCache(cfg.n_layers, b, 64, cfg.n_kv_heads, cfg.head_dim,
                     device="cuda")
    prompt = torch.randint(0, cfg.vocab, (b, 4), device="cuda")
    model.prefill(prompt, kv_ref)
    model.prefill(prompt, kv_meg)
    tok = torch.randint(0, cfg.vocab, (b,), device="cuda")

    meg = MegaQwen3Decode(model, kv_meg, batch=b)
    for step in range(3):
        # eager reference step (torch mode avoids symm ctx shape limits)
        pos = kv_ref.offset.reshape(1, 1).expand(b, 1)
        ref_logits = model.step(tok.view(b, 1), kv_ref, pos, prefill=False,
                                mode="torch")
        kv_ref.advance(1)
        meg_logits = meg.step(tok)
        torch.cuda.synchronize()
        assert_allclose(meg_logits[:, :cfg.vocab], ref_logits,
                        atol=1e-1, rtol=5e-2, msg=f"step {step}")
        assert int(kv_meg.offset) == int(kv_ref.offset)
        tok = ref_logits.argmax(-1)


def test_megakernel_fused_decode_matches_eager():
    """Hop-fused graph (TD_MK_FUSE=1: norm-in-GEMM partials, fused
    prologue+flash-decode, ping-pong residual updates) vs eager."""
    import os

    import triton_dist_amd as td
    from triton_dist_amd.mega import MegaQwen3Decode
    from triton_dist_amd.models import DenseLLM, KVCache, get_config
    from triton_dist_amd.utils import assert_allclose

    td.initialize_distributed()
    td.init_symm_heap()
    cfg = get_config("tiny-gpu", tp_mode="ag_rs", max_length=128)
    model = DenseLLM(cfg, device="cuda")
    model.init_weights(seed=7)
    b = 16  # <= 32 so the fusion gate engages
    model.init_dist_ctx(max_m_total=128)

    kv_ref = KVCache(cfg.n_layers, b, 64, cfg.n_kv_heads, cfg.head_dim,
                     device="cuda")
    kv_meg = KVCache(cfg.n_layers, b, 64, cfg.n_kv_heads, cfg.head_dim,
                     device="cuda")
    prompt = torch.randint(0, cfg.vocab, (b, 4), device="cuda")
    model.prefill(prompt, kv_ref)
    model.prefill(prompt, kv_meg)
    tok = torch.randint(0, cfg.vocab, (b,), device="cuda")

    os.environ["TD_MK_FUSE"] = "1"
    try:
        meg = MegaQwen3Decode(model, kv_meg, batch=b)
    finally:
        del os.environ["TD_MK_FUSE"]
    assert meg.fused
    for step in range(3):
        pos = kv_ref.offset.reshape(1, 1).expand(b, 1)
        ref_logits = model.step(tok.view(b, 1), kv_ref, pos, prefill=False,
                                mode="torch")
        kv_ref.advance(1)
        meg_logits = meg.step(tok)
        torch.cuda.synchronize()
        assert_allclose(meg_logits[:, :cfg.vocab], ref_logits,
                        atol=1e-1, rtol=5e-2, msg=f"step {step}")
        tok = ref_logits.argmax(-1)
