"""Model configs (capability parity with Triton-distributed
python/triton_dist/models/config.py:30-37 ModelConfig + HF-geometry
presets; weights here are random-init — no network in this environment).
"""
from __future__ import annotations

from dataclasses import dataclass, field


@dataclass
class ModelConfig:
    name: str = "qwen3-32b"
    hidden: int = 5120
    intermediate: int = 25600
    n_layers: int = 64
    n_heads: int = 64
    n_kv_heads: int = 8
    head_dim: int = 128
    vocab: int = 151936
    rope_base: float = 1e6
    rms_eps: float = 1e-6
    qk_norm: bool = True
    tie_embeddings: bool = False
    max_length: int = 2048
    dtype: str = "bfloat16"
    tp_mode: str = "ag_rs"  # ag_rs | gemm_ar | allreduce | torch
    # MoE fields (n_experts == 0 -> dense)
    n_experts: int = 0
    moe_topk: int = 8
    # hybrid GDN (Qwen3-Next geometry): 0 = pure attention; N > 0 keeps
    # every N-th layer full attention, the rest are GDN mixers
    gdn_period: int = 0
    gdn_heads: int = 0
    gdn_head_k: int = 128
    gdn_head_v: int = 128
    moe_impl: str = "ep"   # "ep" (expert-parallel) | "tp" (inter-sharded)
    moe_inter: int = 0


PRESETS = {
    # Qwen3-32B geometry (HF Qwen/Qwen3-32B config.json)
    "qwen3-32b": dict(hidden=5120, intermediate=25600, n_layers=64,
                      n_heads=64, n_kv_heads=8, head_dim=128, vocab=151936),
    # Seed-OSS-36B-class geometry (approximate: no HF access here)
    "seed-oss-36b": dict(hidden=5120, intermediate=27648, n_layers=64,
                         n_heads=80, n_kv_heads=8, head_dim=128,
                         vocab=155136),
    # Qwen3-8B geometry
    "qwen3-8b": dict(hidden=4096, intermediate=12288, n_layers=36,
                     n_heads=32, n_kv_heads=8, head_dim=128, vocab=151936),
    # tiny config for CPU tests
    "tiny": dict(hidden=256, intermediate=512, n_layers=2, n_heads=4,
                 n_kv_heads=2, head_dim=64, vocab=512),
    # tiny GPU config: every sharded GEMM dim satisfies the 128/64 tiling
    # at TP in {1, 2}
    "tiny-gpu": dict(hidden=512, intermediate=1024, n_layers=2, n_heads=4,
                     n_kv_heads=2, head_dim=128, vocab=1024),
    # Qwen3-30B-A3B geometry (HF config)
    "qwen3-30b-a3b": dict(hidden=2048, intermediate=6144, n_layers=48,
                          n_heads=32, n_kv_heads=4, head_dim=128,
                          vocab=151936, n_experts=128, moe_topk=8,
                          moe_inter=768),
    # hybrid GDN demo family (Qwen3-Next geometry; random-init only)
    "qwen3-next-like": dict(hidden=2048, intermediate=5120, n_layers=48,
                            n_heads=16, n_kv_heads=2, head_dim=128,
                            vocab=151936, gdn_period=4, gdn_heads=32,
                            gdn_head_k=128, gdn_head_v=128),
    "tiny-gdn": dict(hidden=64, intermediate=128, n_layers=3, n_heads=2,
                     n_kv_heads=2, head_dim=32, vocab=256, gdn_period=3,
                     gdn_heads=4, gdn_head_k=16, gdn_head_v=16),
    "tiny-gdn4": dict(hidden=64, intermediate=128, n_layers=3, n_heads=4,
                      n_kv_heads=4, head_dim=16, vocab=256, gdn_period=3,
                      gdn_heads=4, gdn_head_k=16, gdn_head_v=16),
    # tiny configs divisible at TP=4 (4-rank gloo tests)
    "tiny4": dict(hidden=256, intermediate=512, n_layers=2, n_heads=8,
                  n_kv_heads=4, head_dim=64, vocab=512),
    "tiny-moe4": dict(hidden=64, intermediate=128, n_layers=2, n_heads=4,
                      n_kv_heads=4, head_dim=32, vocab=256, n_experts=4,
                      moe_topk=2, moe_inter=32),
    # tiny MoE for CPU tests
    "tiny-moe": dict(hidden=64, intermediate=128, n_layers=2, n_heads=2,
                     n_kv_heads=2, head_dim=32, vocab=256, n_experts=4,
                     moe_topk=2, moe_inter=32),
    # tiny MoE for GPU tests (128/64-tileable expert GEMMs)
    "tiny-moe-gpu": dict(hidden=512, intermediate=1024, n_layers=2,
                         n_heads=4, n_kv_heads=2, head_dim=128, vocab=1024,
                         n_experts=4, moe_topk=2, moe_inter=128),
}


def get_config(name: str, **overrides) -> ModelConfig:
    cfg = ModelConfig(name=name, **PRESETS[name])
    for k, v in overrides.items():
        setattr(cfg, k, v)
    return cfg
