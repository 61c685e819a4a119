"""CPU tests: Qwen3MoE model — EP ag_rs decode vs torch golden (gloo)."""
import torch

from tests.conftest import run_distributed


def _body_moe_model(rank, world, model_name="tiny-moe"):
    from triton_dist_amd.models import KVCache, Qwen3MoE, get_config
    from triton_dist_amd.utils import assert_allclose

    cfg = get_config(model_name, tp_mode="ag_rs", max_length=64)
    model = Qwen3MoE(cfg, device="cpu")
    model.init_weights(seed=3)
    b, s = world * 2, 4
    model.init_dist_ctx(max_m_total=b)
    kvh = cfg.n_kv_heads // world
    kv1 = KVCache(cfg.n_layers, b, 32, kvh, cfg.head_dim)
    kv2 = KVCache(cfg.n_layers, b, 32, kvh, cfg.head_dim)
    tokens = torch.randint(0, cfg.vocab, (b, s),
                           generator=torch.Generator().manual_seed(1))
    first1 = model.prefill(tokens, kv1)
    first2 = model.prefill(tokens, kv2)
    assert torch.equal(first1, first2)
    pos = kv1.offset.reshape(1, 1).expand(b, 1)
    logits_dist = model.step(first1.view(b, 1), kv1, pos, prefill=False)
    pos2 = kv2.offset.reshape(1, 1).expand(b, 1)
    logits_ref = model.step(first2.view(b, 1), kv2, pos2, prefill=False,
                            mode="torch")
    assert_allclose(logits_dist, logits_ref, atol=1e-1, rtol=5e-2)


def test_moe_model_cpu_2rank():
    run_distributed(_body_moe_model, world_size=2)


def _body_moe_model4(rank, world):
    _body_moe_model(rank, world, model_name="tiny-moe4")


def test_moe_model_cpu_4rank():
    run_distributed(_body_moe_model4, world_size=4)
