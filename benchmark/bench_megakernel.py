#!/usr/bin/env python3
"""Megakernel vs hipGraph-captured layer-op decode (single GPU) — the
reference's headline megakernel table (megakernel.md:29-41: Qwen3 decode
bsz=1 ctx=512, eager/graph/megakernel).

Run: python benchmark/bench_megakernel.py [--model qwen3-8b] [--batch 1]
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="qwen3-8b")
    p.add_argument("--layers", type=int, default=0)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--ctx", type=int, default=512)
    p.add_argument("--iters", type=int, default=20)
    args = p.parse_args()

    import triton_dist_amd as td
    from triton_dist_amd.mega import MegaQwen3Decode
    from triton_dist_amd.models import DenseLLM, Engine, KVCache, get_config

    td.initialize_distributed()
    td.init_symm_heap()
    over = {"tp_mode": "ag_rs", "max_length": args.ctx + 128}
    if args.layers:
        over["n_layers"] = args.layers
    cfg = get_config(args.model, **over)
    model = DenseLLM(cfg, device="cuda")
    model.init_weights(seed=1)
    b = args.batch
    maxlen = args.ctx + args.iters + 16

    # mega path
    kv_m = KVCache(cfg.n_layers, b, maxlen, cfg.n_kv_heads, cfg.head_dim,
                   device="cuda")
    kv_m.offset.fill_(args.ctx)
    meg = MegaQwen3Decode(model, kv_m, batch=b)
    tok = torch.randint(0, cfg.vocab, (b,), device="cuda")
    for _ in range(5):
        meg.step(tok)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        meg.step(tok)
    torch.cuda.synchronize()
    t_mega = (time.perf_counter() - t0) / args.iters * 1e3

    # graph-captured layer-op path (batch padded to the GEMM tiling)
    bg = max(b, 128)
    model.init_dist_ctx(max_m_total=bg)
    eng = Engine(model, batch=bg, max_len=maxlen, use_graph=True)
    eng.kv.offset.fill_(args.ctx)
    eng._ensure_graph()
    eng._token_buf.copy_(torch.randint(0, cfg.vocab, (bg,), device="cuda"))
    for _ in range(5):
        eng.graph.replay()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        eng.graph.replay()
    torch.cuda.synchronize()
    t_graph = (time.perf_counter() - t0) / args.iters * 1e3

    print(f"{args.model} decode bsz={b} ctx={args.ctx}: megakernel "
          f"{t_mega:.3f} ms ({meg.n_tasks} tasks, 1 launch) | graph+ops "
          f"{t_graph:.3f} ms (batch padded to {bg})")
    td.finalize_distributed()


if __name__ == "__main__":
    main()
