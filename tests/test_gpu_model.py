"""GPU model tests: tiny-gpu DenseLLM through the HIP fused ops, 2 ranks
sharing one GPU; dist path vs torch golden logits; hipGraph decode equals
eager decode."""
import pytest
import torch

from tests.conftest import run_distributed

pytestmark = pytest.mark.gpu


def _body_model_dist_vs_torch(rank, world):
    from triton_dist_amd.models import DenseLLM, KVCache, get_config
    from triton_dist_amd.utils import assert_allclose

    cfg = get_config("tiny-gpu", tp_mode="ag_rs", max_length=128)
    model = DenseLLM(cfg, device="cuda")
    model.init_weights(seed=3)
    b = 128 * world  # per-rank shard of 128 rows for the 128-tile GEMM
    model.init_dist_ctx(max_m_total=b)
    kvh = cfg.n_kv_heads // world

    kv1 = KVCache(cfg.n_layers, b, 64, kvh, cfg.head_dim, device="cuda")
    kv2 = KVCache(cfg.n_layers, b, 64, kvh, cfg.head_dim, device="cuda")
    tokens = torch.randint(0, cfg.vocab, (b, 4), device="cuda",
                           generator=torch.Generator("cuda").manual_seed(1))
    first1 = model.prefill(tokens, kv1)
    first2 = model.prefill(tokens, kv2)
    assert torch.equal(first1, first2)

    pos = kv1.offset.reshape(1, 1).expand(b, 1)
    logits_dist = model.step(first1.view(b, 1), kv1, pos, prefill=False)
    torch.cuda.synchronize()
    pos2 = kv2.offset.reshape(1, 1).expand(b, 1)
    logits_ref = model.step(first2.view(b, 1), kv2, pos2, prefill=False,
                            mode="torch")
    torch.cuda.synchronize()
    assert_allclose(logits_dist, logits_ref, atol=1e-1, rtol=5e-2)


def test_model_dist_vs_torch_2rank_gpu():
    run_distributed(_body_model_dist_vs_torch, world_size=2)


def _body_engine_graph(rank, world):
    from triton_dist_amd.models import DenseLLM, Engine, get_config

    cfg = get_config("tiny-gpu", tp_mode="ag_rs", max_length=128)
    b, s, gen = 128 * world, 4, 6
    prompt = torch.randint(0, cfg.vocab, (b, s), device="cuda",
                           generator=torch.Generator("cuda").manual_seed(2))

    model = DenseLLM(cfg, device="cuda")
    model.init_weights(seed=4)
    model.init_dist_ctx(max_m_total=b)

    eng_eager = Engine(model, batch=b, max_len=64, use_graph=False)
    out_eager = eng_eager.serve(prompt, gen_len=gen)
    torch.cuda.synchronize()

    eng_graph = Engine(model, batch=b, max_len=64, use_graph=True)
    out_graph = eng_graph.serve(prompt, gen_len=gen)
    torch.cuda.synchronize()

    match = (out_eager == out_graph).float().mean().item()
    assert match > 0.95, f"graph vs eager token match only {match:.3f}"


def test_engine_graph_2rank_gpu():
    run_distributed(_body_engine_graph, world_size=2)
