"""GPU Qwen3MoE model test: 2 ranks sharing a GPU, EP decode vs torch."""
import pytest
import torch

from tests.conftest import run_distributed

pytestmark = pytest.mark.gpu


def _body_moe_model_gpu(rank, world):
    from triton_dist_amd.models import KVCache, Qwen3MoE, get_config
    from triton_dist_amd.utils import assert_allclose

    cfg = get_config("tiny-moe-gpu", tp_mode="ag_rs", max_length=128)
    model = Qwen3MoE(cfg, device="cuda")
    model.init_weights(seed=3)
    b = 128 * world
    model.init_dist_ctx(max_m_total=b)
    kvh = cfg.n_kv_heads // world
    kv1 = KVCache(cfg.n_layers, b, 64, kvh, cfg.head_dim, device="cuda")
    kv2 = KVCache(cfg.n_layers, b, 64, kvh, cfg.head_dim, device="cuda")
    tokens = torch.randint(0, cfg.vocab, (b, 4), device="cuda",
                           generator=torch.Generator("cuda").manual_seed(1))
    first1 = model.prefill(tokens, kv1)
    first2 = model.prefill(tokens, kv2)
    pos = kv1.offset.reshape(1, 1).expand(b, 1)
    logits_dist = model.step(first1.view(b, 1), kv1, pos, prefill=False)
    torch.cuda.synchronize()
    pos2 = kv2.offset.reshape(1, 1).expand(b, 1)
    logits_ref = model.step(first2.view(b, 1), kv2, pos2, prefill=False,
                            mode="torch")
    torch.cuda.synchronize()
    assert_allclose(logits_dist, logits_ref, atol=1e-1, rtol=5e-2)


def test_moe_model_gpu_2rank():
    run_distributed(_body_moe_model_gpu, world_size=2)
