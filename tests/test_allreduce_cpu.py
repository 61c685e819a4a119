"""CPU tests of the AllReduce op + gemm_ar layer mode (gloo + shm mock)."""
import torch

from tests.conftest import run_distributed


def _body_ar(rank, world):
    from triton_dist_amd.ops import all_reduce, all_reduce_ref, create_allreduce_context
    from triton_dist_amd.utils import assert_allclose, rand_tensor

    ctx = create_allreduce_context(max_elems=4096)
    g = torch.Generator().manual_seed(3 + rank)
    x = rand_tensor((64, 64), dtype=torch.bfloat16, generator=g)
    for _ in range(3):
        out = all_reduce(x, ctx)
        ref = all_reduce_ref(x)
        assert_allclose(out, ref, atol=5e-2, rtol=5e-2)


def test_allreduce_cpu_2rank():
    run_distributed(_body_ar, world_size=2)


def test_allreduce_cpu_4rank():
    run_distributed(_body_ar, world_size=4)


def _body_model_gemm_ar(rank, world):
    from triton_dist_amd.models import DenseLLM, KVCache, get_config
    from triton_dist_amd.utils import assert_allclose

    cfg = get_config("tiny", tp_mode="gemm_ar", max_length=64)
    model = DenseLLM(cfg, device="cpu")
    model.init_weights(seed=3)
    b, s = 4, 4
    model.init_dist_ctx(max_m_total=b)
    kvh = cfg.n_kv_heads // world
    kv1 = KVCache(cfg.n_layers, b, 32, kvh, cfg.head_dim)
    kv2 = KVCache(cfg.n_layers, b, 32, kvh, cfg.head_dim)
    tokens = torch.randint(0, cfg.vocab, (b, s),
                           generator=torch.Generator().manual_seed(1))
    first1 = model.prefill(tokens, kv1)
    first2 = model.prefill(tokens, kv2)
    pos = kv1.offset.reshape(1, 1).expand(b, 1)
    logits_dist = model.step(first1.view(b, 1), kv1, pos, prefill=False)
    pos2 = kv2.offset.reshape(1, 1).expand(b, 1)
    logits_ref = model.step(first2.view(b, 1), kv2, pos2, prefill=False,
                            mode="torch")
    assert_allclose(logits_dist, logits_ref, atol=1e-1, rtol=5e-2)


def test_model_gemm_ar_cpu_2rank():
    run_distributed(_body_model_gemm_ar, world_size=2)


def _body_gemm_ar_tiled(rank, world):
    from triton_dist_amd.ops import create_allreduce_context, gemm_allreduce
    from triton_dist_amd.ops.allreduce import _n_owned
    from triton_dist_amd.utils import assert_allclose, rand_tensor

    # m=512, n=256 -> 2 tiles round-robin across ranks; k=128 hits the
    # fused tile path's shape gate
    ctx = create_allreduce_context(max_elems=512 * 256)
    assert ctx.tile_scatter is not None
    g = torch.Generator().manual_seed(11 + rank)
    a = rand_tensor((512, 128), dtype=torch.bfloat16, generator=g) / 8
    w = rand_tensor((256, 128), dtype=torch.bfloat16, generator=g) / 8
    import torch.distributed as dist
    for _ in range(2):  # back-to-back calls reuse the tile buffers
        outd = gemm_allreduce(a, w, ctx)
        ref = (a.float() @ w.float().t())
        dist.all_reduce(ref)
        assert_allclose(outd, ref.to(torch.bfloat16), atol=8e-2, rtol=5e-2)
    # owner coverage arithmetic
    tiles = (512 // 256) * (256 // 256)
    assert sum(_n_owned(tiles, r, world) for r in range(world)) == tiles


def test_gemm_ar_tiled_cpu_2rank():
    run_distributed(_body_gemm_ar_tiled, world_size=2)


def test_gemm_ar_tiled_cpu_4rank():
    run_distributed(_body_gemm_ar_tiled, world_size=4)
