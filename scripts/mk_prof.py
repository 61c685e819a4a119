"""Per-task-type wait/body attribution of the megakernel (TD_MK_PROF=1)."""
import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
os.environ.setdefault("TD_MK_PROF", "1")

import torch


def main():
    import triton_dist_amd as td
    from triton_dist_amd.mega import MegaQwen3Decode
    from triton_dist_amd.models import DenseLLM, KVCache, get_config

    td.initialize_distributed()
    td.init_symm_heap()
    cfg = get_config("qwen3-8b", tp_mode="ag_rs", max_length=640)
    model = DenseLLM(cfg, device="cuda")
    model.init_weights(seed=1)
    kv = KVCache(cfg.n_layers, 1, 600, cfg.n_kv_heads, cfg.head_dim,
                 device="cuda")
    kv.offset.fill_(512)
    meg = MegaQwen3Decode(model, kv, batch=1)
    tok = torch.randint(0, cfg.vocab, (1,), device="cuda")
    for _ in range(3):
        meg.step(tok)
    torch.cuda.synchronize()
    meg.run.prof.zero_()
    steps = 5
    import time
    t0 = time.perf_counter()
    for _ in range(steps):
        meg.step(tok)
    torch.cuda.synchronize()
    wall = (time.perf_counter() - t0) / steps * 1e3
    prof = meg.run.prof.cpu()
    names = ["RMSNORM", "ADD_RMSNORM", "GEMM_TILE", "SWIGLU",
             "QKV_PROLOGUE", "FLASH_DECODE", "EMBED", "KV_ADVANCE",
             "GEMM_PART", "TILE_REDUCE", "GEMV", "?11", "?12", "?13",
             "?14", "?15"]
    total = prof.sum().item()
    print(f"wall {wall:.2f} ms/step; accumulated WG-ticks {total} "
          f"(100 MHz, {steps} steps, thread0 view)")
    for i, nm in enumerate(names):
        w, b = prof[i, 0].item(), prof[i, 1].item()
        if w + b == 0:
            continue
        print(f"{nm:14s} wait {w/1e5/steps:8.2f} ms  body"
              f" {b/1e5/steps:8.2f} ms  ({(w+b)/max(total,1)*100:5.1f}%)")


if __name__ == "__main__":
    main()
