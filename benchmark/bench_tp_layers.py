"""TP layer benchmarks: TP_MLP and TP_Attn forward latency per mode
(ag_rs / gemm_ar / allreduce / torch), prefill- and decode-shaped
(reference parity: benchmark/bench_tp_mlp.py + bench_tp_attn.py —
behavior only).

Run: bash scripts/launch.sh 8 benchmark/bench_tp_layers.py \\
         [--hidden 5120 --inter 25600 --m 512]
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist

import triton_dist_amd as td
from triton_dist_amd.layers.tp_attn import TP_Attn
from triton_dist_amd.layers.tp_mlp import TP_MLP
from triton_dist_amd.layers.norm import Rotary


def timeit(fn, iters=20, warmup=3):
    for _ in range(warmup):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    t = torch.tensor([us])
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--hidden", type=int, default=5120)
    ap.add_argument("--inter", type=int, default=25600)
    ap.add_argument("--heads", type=int, default=64)
    ap.add_argument("--kv-heads", type=int, default=8)
    ap.add_argument("--tokens", type=int, default=512)  # per rank
    ap.add_argument("--modes", default="ag_rs,allreduce")
    ap.add_argument("--check", action="store_true")
    args = ap.parse_args()
    td.initialize_distributed()
    heap = td.init_symm_heap(
        size_mb=max(1024, 4 * args.tokens * 8 * args.hidden * 2 // int(1e6)))
    world, rank = heap.world, heap.rank
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    m_total = args.tokens * world

    for mode in args.modes.split(","):
        mlp = TP_MLP(args.hidden, args.inter, mode=mode, device=dev)
        mlp.init_weights(seed=1)
        mlp.init_ctx(max_m_total=m_total)
        x = (torch.randn(args.tokens, args.hidden, device=dev) / 8).to(
            torch.bfloat16)
        if args.check and mode == "ag_rs":
            # gather the shard stream and compare vs the replicated golden
            # stage via CPU: routes to gloo under both the gloo-only
            # (ranks sharing one GPU) and cpu:gloo,cuda:nccl groups
            x_cpu = x.cpu()
            xs = [torch.empty_like(x_cpu) for _ in range(world)]
            dist.all_gather(xs, x_cpu)
            full = torch.cat([t.to(dev) for t in xs])
            y = mlp(x)
            ref = mlp.torch_fwd(full)[rank * args.tokens:
                                      (rank + 1) * args.tokens]
            err = (y.float() - ref.float()).abs().max().item()
            if rank == 0:
                print(f"TP_MLP {mode} check: max err {err:.3f} "
                      f"({'OK' if err < 0.5 else 'FAIL'})")
        us = timeit(lambda: mlp(x))
        gf = 2 * m_total * 3 * args.hidden * args.inter / world / 1e9
        if rank == 0:
            print(f"TP_MLP  {mode:9s} m/rank={args.tokens}: {us:9.1f} us "
                  f"({gf/us*1e3:6.0f} TF/rank)")

        head_dim = args.hidden // args.heads * 2  # GQA-ish tiny default
        head_dim = 128 if args.hidden >= 1024 else head_dim
        rot = Rotary(head_dim, 4096, device=dev)
        attn = TP_Attn(args.hidden, args.heads, args.kv_heads, head_dim,
                       rot, mode=mode, device=dev)
        attn.init_weights(seed=2)
        attn.init_ctx(max_m_total=m_total)
        # ag_rs shards the batch (rows gather to m_total); the replicated
        # modes see the local batch as the full batch
        b_attn = m_total if mode == "ag_rs" else args.tokens
        pos = torch.zeros(b_attn, 1, dtype=torch.int64, device=dev)
        us = timeit(lambda: attn(x, None, 0, pos, b_attn, 1, True))
        if rank == 0:
            print(f"TP_Attn {mode:9s} m/rank={args.tokens}: {us:9.1f} us")
    td.shutdown_heap()


if __name__ == "__main__":
    main()
