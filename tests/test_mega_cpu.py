"""CPU tests for the megakernel BUILDER: task emission, scoreboard deps,
and the opt-in K-split decomposition (TD_MK_KSPLIT)."""
import torch

from triton_dist_amd.mega.builder import (MegaGraph, T_GEMM_TILE,
                                          T_GEMM_TILE_PART, T_TILE_REDUCE,
                                          emit_gemm)


def test_emit_gemm_plain():
    g = MegaGraph()
    dep = g.new_op()
    g.next_level()
    op = emit_gemm(g, 1, 2, 3, batch=64, n=256, k=512, dep=dep)
    tiles = (64 // 32) * (256 // 128)
    assert op.n_tasks == tiles
    assert all(t[0] == T_GEMM_TILE for t in g.tasks)
    # every task depends on `dep` with its full count
    for t in g.tasks:
        assert t[2] == dep.slot


def test_emit_gemm_ksplit():
    g = MegaGraph()
    op = emit_gemm(g, 1, 2, 3, batch=32, n=256, k=512, dep=None, ksplit=4,
                   ws_ptr=99)
    parts = [t for t in g.tasks if t[0] == T_GEMM_TILE_PART]
    reds = [t for t in g.tasks if t[0] == T_TILE_REDUCE]
    tiles = 1 * 2
    assert len(parts) == tiles * 4 and len(reds) == tiles
    # K ranges tile the full K
    ranges = sorted((t[6][8], t[6][9]) for t in parts if t[6][7] == 0
                    and t[6][6] == 0)
    assert ranges == [(0, 128), (128, 128), (256, 128), (384, 128)]
    # reduce waits the parts op at FULL count, consumer op is the reduce
    parts_slot = parts[0][1]
    for t in reds:
        assert t[2] == parts_slot and t[3] == len(parts)
    assert op.slot == reds[0][1]
    # parts at an earlier level than reduces
    assert max(t[7] for t in parts) < min(t[7] for t in reds)


def test_emit_gemm_ksplit_indivisible_falls_back():
    g = MegaGraph()
    op = emit_gemm(g, 1, 2, 3, batch=32, n=128, k=192, dep=None, ksplit=4,
                   ws_ptr=99)  # 192 % 256 != 0
    assert all(t[0] == T_GEMM_TILE for t in g.tasks)
    assert op.n_tasks == 1


def test_finalize_shapes():
    g = MegaGraph()
    emit_gemm(g, 1, 2, 3, batch=32, n=256, k=512, dep=None, ksplit=2,
              ws_ptr=9)
    buf, queue, offs, score = g.finalize(n_wg=4, device="cpu")
    assert buf.shape[0] == len(g.tasks) and queue.numel() == len(g.tasks)
    assert offs[-1].item() == len(g.tasks)
    assert score.numel() == len(g.ops)


def test_ksplit_task_semantics():
    """Execute emit_gemm's K-split task stream in numpy (each
    T_GEMM_TILE_PART computes its (tile, k-range) partial into its ws
    slice; each T_TILE_REDUCE sums slices) and compare against the full
    matmul — validates the task-arg encoding and the splitting math
    end-to-end on CPU."""
    import numpy as np

    BM, BN = 32, 128
    batch, n, k, ksplit = 70, 256, 512, 4   # batch NOT a tile multiple
    m_pad = (batch + BM - 1) // BM * BM
    rng = np.random.default_rng(0)
    A = rng.standard_normal((m_pad, k)).astype(np.float32)
    B = rng.standard_normal((n, k)).astype(np.float32)
    C = np.zeros((batch, n), dtype=np.float32)
    ws = np.zeros((ksplit, m_pad, n), dtype=np.float32)

    g = MegaGraph()
    emit_gemm(g, 1, 2, 3, batch=batch, n=n, k=k, dep=None, ksplit=ksplit,
              ws_ptr=9)
    for (tt, _slot, _d0, _d0n, _d1, _d1n, args, _lvl) in g.tasks:
        if tt == T_GEMM_TILE_PART:
            (_a, _b, _w, m, nn, kk, pm, pn, k0, klen, sk) = args[:11]
            rows = min(BM, m - pm * BM)
            a_blk = A[pm * BM:pm * BM + rows, k0:k0 + klen]
            b_blk = B[pn * BN:(pn + 1) * BN, k0:k0 + klen]
            ws[sk, pm * BM:pm * BM + rows, pn * BN:(pn + 1) * BN] = \
                a_blk @ b_blk.T
        elif tt == T_TILE_REDUCE:
            (_w, _c, m, nn, pm, pn, ks) = args[:7]
            rows = min(BM, m - pm * BM)
            C[pm * BM:pm * BM + rows, pn * BN:(pn + 1) * BN] = \
                ws[:ks, pm * BM:pm * BM + rows,
                   pn * BN:(pn + 1) * BN].sum(0)
    ref = A[:batch] @ B.T
    assert np.abs(C - ref).max() < 1e-3


def test_emit_gemv_semantics():
    """Execute emit_gemv's task stream in numpy: every 512-col chunk task
    computes C[:, c0:c0+chunk] = A @ W[c0:c0+chunk].T; together they must
    reproduce the full matmul, covering every column exactly once."""
    import numpy as np

    from triton_dist_amd.mega.builder import MegaGraph, T_GEMV, emit_gemv

    rng = np.random.default_rng(3)
    for batch, n, k in [(1, 1536, 512), (2, 1024, 256), (4, 2368, 128)]:
        g = MegaGraph()
        A = rng.standard_normal((batch, k)).astype(np.float32)
        W = rng.standard_normal((n, k)).astype(np.float32)
        C = np.zeros((batch, n), dtype=np.float32)
        emit_gemv(g, 0, 0, 0, batch, n, k, None)
        cols = []
        for t in g.tasks:
            tt, args = t[0], t[6]
            assert tt == T_GEMV
            _, _, _, m, nn, kk, c0, chunk = args[:8]
            assert (m, nn, kk) == (batch, n, k)
            C[:, c0:c0 + chunk] = A @ W[c0:c0 + chunk].T
            cols += list(range(c0, c0 + chunk))
        assert sorted(cols) == list(range(n))
        np.testing.assert_allclose(C, A @ W.T, rtol=1e-4, atol=1e-3)


def test_fused_graph_structure():
    """TD_MK_FUSE=1: the three per-layer serialization hops disappear —
    no standalone rmsnorm / prologue / flash-decode tasks; norm-fused
    GEMM partials and fused prologue+decode tasks appear instead, and
    every dep slot points at a valid op."""
    import os

    import torch

    from triton_dist_amd.mega.builder import (
        T_ADD_RMSNORM, T_FLASH_DECODE, T_GEMM_TILE_PART,
        T_GEMM_TILE_PART_NR, T_PRO_FLASH_DECODE, T_QKV_PROLOGUE, T_RMSNORM)
    from triton_dist_amd.mega.qwen3 import MegaQwen3Decode
    from triton_dist_amd.models import DenseLLM, KVCache, get_config

    cfg = get_config("tiny-gpu", max_length=64)
    model = DenseLLM(cfg, device="cpu")
    model.init_weights(seed=3)
    kv = KVCache(cfg.n_layers, 8, 64, cfg.n_kv_heads, cfg.head_dim,
                 device="cpu")

    def types(meg):
        return (meg.run.task_buf[:, 0] & 0xFFFFFFFF).tolist()

    os.environ["TD_MK_FUSE"] = "1"
    try:
        fused = MegaQwen3Decode(model, kv, batch=8)
    finally:
        del os.environ["TD_MK_FUSE"]
    assert fused.fused
    tf = types(fused)
    assert tf.count(T_RMSNORM) == 0
    assert tf.count(T_QKV_PROLOGUE) == 0
    assert tf.count(T_FLASH_DECODE) == 0
    # one fused prologue+decode per (b, kh) per layer
    assert tf.count(T_PRO_FLASH_DECODE) == 8 * cfg.n_kv_heads * cfg.n_layers
    # qkv + gate_up partials are norm-fused; o/down/lm stay plain
    assert tf.count(T_GEMM_TILE_PART_NR) > 0
    assert tf.count(T_GEMM_TILE_PART) > 0
    # residual updates ride parallel ADD_RMSNORM tasks (ping-pong x)
    assert tf.count(T_ADD_RMSNORM) > 0
    # dep slots reference valid ops
    n_ops = fused.run.n_ops
    d0 = (fused.run.task_buf[:, 1] & 0xFFFFFFFF).to(torch.int32)
    d1 = (fused.run.task_buf[:, 2] & 0xFFFFFFFF).to(torch.int32)
    assert int(d0.max()) < n_ops and int(d1.max()) < n_ops

    kv2 = KVCache(cfg.n_layers, 8, 64, cfg.n_kv_heads, cfg.head_dim,
                  device="cpu")
    plain = MegaQwen3Decode(model, kv2, batch=8)
    assert not plain.fused
    tp = types(plain)
    assert tp.count(T_PRO_FLASH_DECODE) == 0
    assert tp.count(T_GEMM_TILE_PART_NR) == 0
    assert tp.count(T_FLASH_DECODE) == 8 * cfg.n_kv_heads * cfg.n_layers


import pytest


@pytest.mark.parametrize("env", [{}, {"TD_MK_FUSE": "1"},
                                 {"TD_MK_PIPE": "1"}])
def test_graph_levels_monotone_with_deps(env, monkeypatch):
    """Scheduler deadlock-freedom precondition: queues execute in level
    order, so every task's dependencies must be PRODUCED at strictly
    lower levels than the task's own level — for the default, hop-fused
    and pipelined emission modes."""
    import os

    import torch

    from triton_dist_amd.mega import builder as B
    from triton_dist_amd.mega.qwen3 import MegaQwen3Decode
    from triton_dist_amd.models import DenseLLM, KVCache, get_config

    captured = {}
    orig = B.MegaGraph.finalize

    def capture(self, n_wg, device="cuda"):
        captured["tasks"] = list(self.tasks)
        captured["n_ops"] = len(self.ops)
        return orig(self, n_wg, device)

    monkeypatch.setattr(B.MegaGraph, "finalize", capture)
    for k, v in env.items():
        monkeypatch.setenv(k, v)
    batch = 1 if env.get("TD_MK_PIPE") else 8
    cfg = get_config("tiny-gpu", max_length=64)
    model = DenseLLM(cfg, device="cpu")
    model.init_weights(seed=2)
    kv = KVCache(cfg.n_layers, batch, 64, cfg.n_kv_heads, cfg.head_dim,
                 device="cpu")
    MegaQwen3Decode(model, kv, batch=batch)
    tasks = captured["tasks"]
    assert tasks
    # per-op max level of its producing tasks
    op_level = {}
    for (tt, slot, d0, d0n, d1, d1n, args, lvl) in tasks:
        op_level[slot] = max(op_level.get(slot, -1), lvl)
    for (tt, slot, d0, d0n, d1, d1n, args, lvl) in tasks:
        for d in (d0, d1):
            if d >= 0:
                assert op_level[d] < lvl, (env, tt, slot, d, op_level[d],
                                           lvl)
