"""The tutorial's code paths, executed (reference parity: tutorials 01-10
double as tests). Each section of docs/TUTORIAL.md that can run on CPU
(gloo + shm mock heap) is exercised here so the doc can't rot."""
import torch

from tests.conftest import run_distributed


def _worker_t1_symm(rank, world):
    # Tutorial §1: symmetric memory + notify/wait
    import triton_dist_amd as td
    from triton_dist_amd.runtime import cpu_shm

    heap = td.init_symm_heap(size_mb=16)
    buf = heap.alloc_buffer((4, 16), torch.float32)
    flags = heap.alloc_buffer((8,), torch.int32)
    if rank == 0:
        buf.peer(1)[0].fill_(42.0)
        cpu_shm.notify(flags.peer(1), 0, 1)
    elif rank == 1:
        cpu_shm.wait_ge(flags.local(), 0, 1)
        assert (buf.local()[0] == 42.0).all()
    heap.barrier_all()
    td.shutdown_heap()


def _worker_t2_t3(rank, world):
    # Tutorial §2 (AG-GEMM) + §3 (TP layers)
    import triton_dist_amd as td
    from triton_dist_amd.layers.tp_mlp import TP_MLP
    from triton_dist_amd.ops import ag_gemm, create_ag_gemm_context

    td.init_symm_heap(size_mb=32)
    ctx = create_ag_gemm_context(max_m_per_rank=16, k=32)
    g = torch.Generator().manual_seed(1)
    a = (torch.randn(16, 32, generator=g) / 4).to(torch.bfloat16)
    w = (torch.randn(24, 32, generator=g) / 4).to(torch.bfloat16)
    c = ag_gemm(a, w, ctx)
    assert c.shape == (world * 16, 24)

    mlp = TP_MLP(hidden=32, intermediate=64, mode="ag_rs")
    mlp.init_ctx(max_m_total=world * 16)
    mlp.init_weights(seed=2)
    y = mlp(a)
    assert y.shape == (16, 32)
    td.shutdown_heap()


def _worker_t4_ep(rank, world):
    # Tutorial §4: EP MoE
    import triton_dist_amd as td
    from triton_dist_amd.ops import create_ep_context, ep_moe_forward

    td.init_symm_heap(size_mb=32)
    H, E, K, I = 32, 4, 2, 16
    ctx = create_ep_context(max_tokens=8, hidden=H, n_experts=E, topk=K)
    g = torch.Generator().manual_seed(3)
    x = (torch.randn(8, H, generator=g) / 4).to(torch.bfloat16)
    ids = torch.randint(0, E, (8, K), generator=g, dtype=torch.int32)
    wts = torch.softmax(torch.randn(8, K, generator=g), -1).float()
    wg = (torch.randn(E // world, 2 * I, H, generator=g) / 4).to(
        torch.bfloat16)
    wd = (torch.randn(E // world, H, I, generator=g) / 4).to(torch.bfloat16)
    y = ep_moe_forward(x, ids, wts, wg, wd, ctx)
    assert y.shape == (8, H)
    td.shutdown_heap()


def _worker_t5_engine(rank, world):
    # Tutorial §5: model + Engine serve
    import triton_dist_amd as td
    from triton_dist_amd.models import AutoLLM, Engine, get_config

    td.init_symm_heap(size_mb=32)
    cfg = get_config("tiny", tp_mode="ag_rs")
    model = AutoLLM(cfg, device="cpu")
    model.init_weights()
    model.init_dist_ctx(max_m_total=world * 4 * 8)
    eng = Engine(model, batch=4, max_len=32, use_graph=False)
    prompt = torch.randint(0, cfg.vocab, (4, 8))
    out = eng.serve(prompt, gen_len=3)
    assert out.shape == (4, 3)
    td.shutdown_heap()


def test_tutorial_symm():
    run_distributed(_worker_t1_symm, world_size=2)


def test_tutorial_ag_gemm_tp():
    run_distributed(_worker_t2_t3, world_size=2)


def test_tutorial_ep_moe():
    run_distributed(_worker_t4_ep, world_size=2)


def test_tutorial_engine():
    run_distributed(_worker_t5_engine, world_size=2)
