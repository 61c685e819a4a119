#!/usr/bin/env python3
"""Flagship benchmark: Qwen3-32B TP=N end-to-end decode (BASELINE config 5).

Contract: `python bench.py --gpus N --steps K --warmup W` — launched for N>1
as one rank per GPU via torch.distributed.run. Does W untimed warmup decode
steps, then times EXACTLY K steps bracketed by barrier+synchronize on both
sides, takes the MAX over ranks, and rank 0 prints ONE JSON line.

Weak scaling: global batch = 512 * N (per-GPU batch fixed at 512), TP=N,
synthetic prompts, random-init weights (no network for checkpoints).
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--model", default="qwen3-32b")
    p.add_argument("--batch-per-gpu", type=int, default=512)
    p.add_argument("--ctx", type=int, default=128)
    p.add_argument("--mode", default="ag_rs",
                   choices=["ag_rs", "gemm_ar", "allreduce", "torch"])
    p.add_argument("--no-graph", action="store_true")
    p.add_argument("--seed", type=int, default=1234)
    return p.parse_args()


def main():
    args = parse_args()
    import torch.distributed as dist

    import triton_dist_amd as td
    from triton_dist_amd.models import AutoLLM, Engine, get_config

    td.initialize_distributed(seed=args.seed)
    world = td.world_size()
    rank = td.rank()
    n_gpus = world if world > 1 else args.gpus
    assert world == n_gpus or world == 1, \
        f"launch with torch.distributed.run for --gpus {args.gpus}"
    on_gpu = torch.cuda.is_available()
    device = "cuda" if on_gpu else "cpu"
    from triton_dist_amd.utils import gpu_oversubscribed
    if on_gpu and args.mode == "ag_rs" and gpu_oversubscribed(world):
        # Oversubscribed DEV topology (ranks REALLY sharing one physical
        # GPU — verified by a PCI-id exchange, so a launcher that pins
        # one visible device per rank at world 8 does NOT trip this):
        # the fused persistent spin-wait consumers of both ranks saturate
        # the single GPU's CUs and the peer's flag-producing kernels
        # starve. One rank per GPU (the benchmark topology) has no such
        # cycle: every spin's producer is SDMA (CU-free) or
        # stream-ordered before the waiter.
        print("[bench] world > device_count: oversubscribed sharing "
              "topology -> falling back to --mode allreduce --no-graph",
              file=sys.stderr)
        args.mode = "allreduce"
        args.no_graph = True  # host collectives are not graph-capturable
    batch = args.batch_per_gpu * world
    max_len = args.ctx + args.warmup + args.steps + 16

    cfg = get_config(args.model, tp_mode=args.mode, max_length=max_len + 64)
    # size the symmetric heap for fused-path PREFILL (M = batch*ctx tokens):
    # AG workspace + RS scatter are each batch*ctx*hidden*2 bytes
    prefill_m = batch * args.ctx
    # MoE models keep contexts at decode size (EP prefill buffers scale
    # with M*topk; torch prefill has no collectives at world=1)
    dense = cfg.n_experts == 0
    # ag_rs sizes contexts for the FUSED prefill (batch*ctx tokens);
    # gemm_ar prefills via the torch path (replicated + RCCL AR), so its
    # symmetric AR contexts only need the decode batch — prefill-sized
    # replicated inboxes (world * M * H) would not fit the heap
    ctx_m = prefill_m if (on_gpu and dense and args.mode == "ag_rs") \
        else batch
    if args.mode == "ag_rs" and on_gpu:
        need = 2.4 * ctx_m * cfg.hidden * 2
        if not dense:  # EP recv/combine symm buffers
            need += 4.6 * ctx_m * cfg.moe_topk * cfg.hidden * 2 / world
        heap = td.init_symm_heap(size_mb=max(int(need / 1e6) + 1024, 4096))
    elif args.mode == "gemm_ar" and on_gpu:
        # one-/two-shot AR inbox+outbox per layer-shared ctx: world
        # replicated copies of the decode activation
        need = 3.0 * world * batch * cfg.hidden * 2
        heap = td.init_symm_heap(size_mb=max(int(need / 1e6) + 1024, 4096))
    else:
        heap = td.init_symm_heap()

    if on_gpu:
        # shipped hipBLASLt/rocBLAS algo picks for the matmul-routed
        # GEMMs (read-only TunableOp file; +6-7% on the qwen3-32b step)
        from triton_dist_amd.tune import maybe_enable_tunableop
        maybe_enable_tunableop()
    model = AutoLLM(cfg, device=device)
    model.init_weights(seed=args.seed)
    if args.mode in ("ag_rs", "gemm_ar"):
        model.init_dist_ctx(max_m_total=ctx_m)

    eng = Engine(model, batch=batch, max_len=max_len,
                 use_graph=on_gpu and not args.no_graph)

    # synthetic prompt + prefill (untimed)
    g = torch.Generator().manual_seed(args.seed)
    prompt = torch.randint(0, cfg.vocab, (batch, args.ctx), generator=g
                           ).to(device)
    eng.kv.reset()
    tok = model.prefill(prompt, eng.kv)

    # warmup decode steps (untimed)
    if eng.use_graph:
        eng._ensure_graph()
        eng._token_buf.copy_(tok)
        for _ in range(args.warmup):
            eng.graph.replay()
    else:
        for _ in range(args.warmup):
            tok = eng.decode_once(tok)

    # timed region
    if dist.is_initialized():
        dist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    if eng.use_graph:
        for _ in range(args.steps):
            eng.graph.replay()
    else:
        for _ in range(args.steps):
            tok = eng.decode_once(tok)
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if dist.is_initialized():
        dist.barrier()
        t = torch.tensor([elapsed])
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed * 1e3 / args.steps
    tok_per_s = batch * args.steps / elapsed
    if rank == 0:
        print(json.dumps({
            "metric": f"{args.model.replace('-', '_')}_tp_decode_tokens_per_s",
            "value": round(tok_per_s, 2),
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": batch,
                "seq_len": args.ctx,
                "parallelism": f"tp{n_gpus}",
                "tp_mode": args.mode,
                "graph": eng.use_graph,
            },
        }))
    td.finalize_distributed()


if __name__ == "__main__":
    main()
