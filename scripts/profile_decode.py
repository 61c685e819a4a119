"""Profile the graph-replayed decode step in isolation (no prefill noise).

Builds the flagship model exactly as bench.py does, captures the decode
graph, then torch-profiles N replays and prints per-kernel totals for ONE
step. This is the decision tool for the decode roadmap (docs/ROADMAP.md #1).
"""
import argparse
import json

import torch

import triton_dist_amd as td
from triton_dist_amd.models import AutoLLM, Engine, get_config


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="qwen3-32b")
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--out", default="gpurun_out/decode_prof.json")
    args = ap.parse_args()

    td.initialize_distributed()
    cfg = get_config(args.model, tp_mode="ag_rs")
    dense = cfg.n_experts == 0
    prefill_m = args.batch * 128
    ctx_m = prefill_m if dense else args.batch
    need = 2.4 * ctx_m * cfg.hidden * 2
    if not dense:
        need += 4.6 * ctx_m * cfg.moe_topk * cfg.hidden * 2
    td.init_symm_heap(size_mb=max(int(need / 1e6) + 1024, 4096))
    model = AutoLLM(cfg, device="cuda")
    model.init_weights()
    model.init_dist_ctx(max_m_total=ctx_m)
    eng = Engine(model, batch=args.batch, max_len=256)
    prompt = torch.randint(0, cfg.vocab, (args.batch, 128),
                           device="cuda")
    eng.serve(prompt, gen_len=3)  # prefill + capture + a couple replays

    rows = eng.profile_decode(steps=args.steps)
    total = sum(r["us_per_step"] for r in rows.values())
    top = sorted(rows.items(), key=lambda kv: -kv[1]["us_per_step"])
    print(f"== decode step kernel totals ({args.model}, B={args.batch}) ==")
    print(f"sum of kernel time: {total / 1000:.2f} ms/step")
    for name, r in top[:30]:
        print(f"  {r['us_per_step']:9.1f} us  x{r['calls']:5d}  {name[:100]}")
    with open(args.out, "w") as f:
        json.dump({"model": args.model, "batch": args.batch,
                   "total_us": total, "kernels": rows}, f, indent=1)


if __name__ == "__main__":
    main()
