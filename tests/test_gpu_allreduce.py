"""GPU allreduce tests: 2 ranks sharing a GPU over hipIpc; one-shot and
two-shot vs the torch.distributed golden."""
import pytest
import torch

from tests.conftest import run_distributed

pytestmark = pytest.mark.gpu


def _body_ar_gpu(rank, world):
    from triton_dist_amd.ops import (all_reduce, all_reduce_ref,
                                     create_allreduce_context)
    from triton_dist_amd.utils import assert_allclose

    ctx = create_allreduce_context(max_elems=8 << 20)
    torch.manual_seed(5 + rank)
    # the (1021, 8)/(1022, 16) rows pin the 8-aligned-chunk-boundary fix:
    # ceil(elems/chunks) is NOT naturally 8-aligned for them, so the
    # vectorized tail must not overrun the inbox row (allreduce.hip)
    for method, shape in (("one_shot", (512, 5120)),
                          ("two_shot", (1024, 5120)),
                          ("one_shot", (8, 1024)),
                          ("one_shot", (1021, 8)),
                          ("two_shot", (1022, 16))):
        x = (torch.randn(shape, device="cuda") / 4).to(torch.bfloat16)
        for _ in range(2):
            out = all_reduce(x, ctx, method=method)
            torch.cuda.synchronize()
            ref = all_reduce_ref(x)
            assert_allclose(out, ref, atol=8e-2, rtol=5e-2,
                            msg=f"{method} {shape}")


def test_allreduce_gpu_2rank():
    run_distributed(_body_ar_gpu, world_size=2)


def _body_model_gemm_ar_gpu(rank, world):
    from triton_dist_amd.models import DenseLLM, KVCache, get_config
    from triton_dist_amd.utils import assert_allclose

    cfg = get_config("tiny-gpu", tp_mode="gemm_ar", max_length=128)
    model = DenseLLM(cfg, device="cuda")
    model.init_weights(seed=3)
    b = 128
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


def test_model_gemm_ar_gpu_2rank():
    run_distributed(_body_model_gemm_ar_gpu, world_size=2)


def _body_gemm_ar_tiled_gpu(rank, world):
    from triton_dist_amd.ops import create_allreduce_context, gemm_allreduce
    from triton_dist_amd.utils import assert_allclose
    import torch.distributed as dist

    ctx = create_allreduce_context(max_elems=2048 * 5120)
    torch.manual_seed(17 + rank)
    # (512, 5120, 3456): decode down-proj shard shape -> split-K producer;
    # (2048, 5120, 1024): plain 256^2 producer path; (512, 512, 128):
    # tiles < consumer-grid edge cases
    for m, n, k in ((512, 5120, 3456), (2048, 5120, 1024),
                    (512, 512, 128)):
        a = (torch.randn(m, k, device="cuda") / 8).to(torch.bfloat16)
        w = (torch.randn(n, k, device="cuda") / 8).to(torch.bfloat16)
        for _ in range(2):  # back-to-back: flag reset/reuse protocol
            out = gemm_allreduce(a, w, ctx)
            torch.cuda.synchronize()
            ref = a.float() @ w.float().t()
            dist.all_reduce(ref)
            assert_allclose(out, ref.to(torch.bfloat16), atol=2.5e-1,
                            rtol=5e-2, msg=f"gemm_ar {m}x{n}x{k}")


def test_gemm_ar_tiled_gpu_2rank():
    run_distributed(_body_gemm_ar_tiled_gpu, world_size=2)
