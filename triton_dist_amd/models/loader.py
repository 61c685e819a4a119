"""HF-checkpoint weight loading with TP/EP sharding.

Capability parity with the reference's load path (Triton-distributed
python/triton_dist/models/dense.py:150-167: HF weights loaded + sharded
on CPU, then moved per layer — behavior only). Works fully OFFLINE from
a local directory of safetensors files (`model.safetensors` or a
`model.safetensors.index.json` shard index); no hub access.

Name mapping (Qwen3 / Qwen3-MoE HF layout -> this framework):
  model.embed_tokens.weight            -> embed
  lm_head.weight                       -> lm_head (falls back to tied)
  model.norm.weight                    -> final_norm_w
  model.layers.N.input_layernorm.weight        -> layers[N]["ln1"]
  model.layers.N.post_attention_layernorm.weight -> layers[N]["ln2"]
  .self_attn.{q,k,v}_proj.weight  -> w_qkv (head-sharded rows, [Q;K;V])
  .self_attn.o_proj.weight        -> w_o   (column shard)
  .self_attn.{q,k}_norm.weight    -> q_norm_w / k_norm_w
  .mlp.{gate,up}_proj.weight      -> w_gate_up (intermediate row shard)
  .mlp.down_proj.weight           -> w_down    (intermediate col shard)
  .mlp.gate.weight                -> EPMoELayer.router (replicated)
  .mlp.experts.E.{gate,up}_proj   -> EPMoELayer.w_gate_up[E - lo]
  .mlp.experts.E.down_proj        -> EPMoELayer.w_down[E - lo]
(EP shards experts contiguously: rank r owns [r*e_loc, (r+1)*e_loc).)
"""
from __future__ import annotations

import json
from pathlib import Path
from typing import Dict


class _ShardedCheckpoint:
    """Lazy tensor access over one or many local safetensors files."""

    def __init__(self, path: str):
        self.dir = Path(path)
        idx = self.dir / "model.safetensors.index.json"
        self._file_of: Dict[str, str] = {}
        if idx.exists():
            m = json.loads(idx.read_text())["weight_map"]
            self._file_of = dict(m)
        else:
            files = sorted(self.dir.glob("*.safetensors"))
            if not files:
                raise FileNotFoundError(f"no safetensors under {self.dir}")
            from safetensors import safe_open
            for f in files:
                with safe_open(str(f), framework="pt") as sf:
                    for k in sf.keys():
                        self._file_of[k] = f.name
        self._open = {}

    def keys(self):
        return self._file_of.keys()

    def __contains__(self, name):
        return name in self._file_of

    def get(self, name):
        from safetensors import safe_open
        fname = self._file_of[name]
        if fname not in self._open:
            self._open[fname] = safe_open(str(self.dir / fname),
                                          framework="pt")
        return self._open[fname].get_tensor(name)


def load_hf_weights(model, path: str, strict: bool = True) -> int:
    """Load + shard a local HF Qwen3(-MoE) checkpoint into `model`
    (DenseLLM or Qwen3MoE). Returns the number of checkpoint tensors
    consumed."""
    import torch

    ckpt = _ShardedCheckpoint(path)
    cfg = model.cfg
    w, r = model.world, model.rank
    d = cfg.head_dim
    dt = model.dtype
    dev = model.device
    used = 0

    def take(name):
        nonlocal used
        used += 1
        return ckpt.get(name).to(torch.float32)

    def put(dst, src):
        assert dst.shape == src.shape, (dst.shape, src.shape)
        dst.copy_(src.to(dt).to(dev))

    put(model.embed, take("model.embed_tokens.weight"))
    if "lm_head.weight" in ckpt:
        put(model.lm_head, take("lm_head.weight"))
    else:
        model.lm_head.copy_(model.embed)
    put(model.final_norm_w, take("model.norm.weight"))

    for li, layer in enumerate(model.layers):
        p = f"model.layers.{li}."
        attn, mlp = layer["attn"], layer["mlp"]
        put(layer["ln1"], take(p + "input_layernorm.weight"))
        put(layer["ln2"], take(p + "post_attention_layernorm.weight"))

        if not hasattr(attn, "w_qkv"):  # GDN mixer layer (hybrid family)
            _load_gdn_mixer(attn, ckpt, p, take, put)
            _load_mlp(mlp, ckpt, p, r, take, put)
            continue
        qh, kvh = attn.qh, attn.kvh
        wq = take(p + "self_attn.q_proj.weight")
        wk = take(p + "self_attn.k_proj.weight")
        wv = take(p + "self_attn.v_proj.weight")
        put(attn.w_qkv, torch.cat([
            wq[r * qh * d:(r + 1) * qh * d],
            wk[r * kvh * d:(r + 1) * kvh * d],
            wv[r * kvh * d:(r + 1) * kvh * d]]))
        wo = take(p + "self_attn.o_proj.weight")
        put(attn.w_o, wo[:, r * qh * d:(r + 1) * qh * d].contiguous())
        if cfg.qk_norm:
            qn = p + "self_attn.q_norm.weight"
            if qn in ckpt:
                put(attn.q_norm_w, take(qn))
                put(attn.k_norm_w, take(p + "self_attn.k_norm.weight"))
            elif strict:
                raise KeyError(qn)

        if hasattr(mlp, "inter_shard"):  # dense TP_MLP
            _load_mlp(mlp, ckpt, p, r, take, put)
        else:  # EPMoELayer
            put(mlp.router, take(p + "mlp.gate.weight"))
            lo = r * mlp.e_loc
            for le in range(mlp.e_loc):
                e = lo + le
                ep = p + f"mlp.experts.{e}."
                wg = take(ep + "gate_proj.weight")
                wu = take(ep + "up_proj.weight")
                put(mlp.w_gate_up[le], torch.cat([wg, wu]))
                put(mlp.w_down[le], take(ep + "down_proj.weight"))
    return used


def _save_mlp(out, p, mlp):
    i_s = mlp.inter_shard
    gu = mlp.w_gate_up.cpu()
    out[p + "mlp.gate_proj.weight"] = gu[:i_s]
    out[p + "mlp.up_proj.weight"] = gu[i_s:]
    out[p + "mlp.down_proj.weight"] = mlp.w_down.cpu()


def _load_mlp(mlp, ckpt, p, r, take, put):
    import torch

    i_s = mlp.inter_shard
    wg = take(p + "mlp.gate_proj.weight")
    wu = take(p + "mlp.up_proj.weight")
    put(mlp.w_gate_up,
        torch.cat([wg[r * i_s:(r + 1) * i_s],
                   wu[r * i_s:(r + 1) * i_s]]))
    wd = take(p + "mlp.down_proj.weight")
    put(mlp.w_down, wd[:, r * i_s:(r + 1) * i_s].contiguous())


def _load_gdn_mixer(attn, ckpt, p, take, put):
    """GDN mixer shard: heads split across ranks; checkpoint names use a
    `gdn.` block (our own serialization — no HF Qwen3-Next mapping is
    attempted, this family is random-init/round-trip only)."""
    import torch

    r, lh = attn.rank, attn.lh
    dk, dv = attn.dk, attn.dv
    wq = take(p + "gdn.q_proj.weight")
    wk = take(p + "gdn.k_proj.weight")
    wv = take(p + "gdn.v_proj.weight")
    wg = take(p + "gdn.g_proj.weight")
    wb = take(p + "gdn.b_proj.weight")
    wo = take(p + "gdn.o_proj.weight")
    pad = torch.zeros(attn.proj_dim - attn._proj_used, attn.hidden)
    put(attn.w_in, torch.cat([
        wq[r * lh * dk:(r + 1) * lh * dk],
        wk[r * lh * dk:(r + 1) * lh * dk],
        wv[r * lh * dv:(r + 1) * lh * dv],
        wg[r * lh:(r + 1) * lh],
        wb[r * lh:(r + 1) * lh],
        pad]))
    put(attn.w_out, wo[:, r * lh * dv:(r + 1) * lh * dv].contiguous())


def save_hf_weights(model, path: str):
    """Inverse mapping (world == 1 only): write the model's weights as a
    single local `model.safetensors` in HF Qwen3(-MoE) naming. Used for
    round-trip tests and for exporting random-init checkpoints."""
    import torch
    from safetensors.torch import save_file

    assert model.world == 1, "save_hf_weights: world must be 1"
    cfg = model.cfg
    d = cfg.head_dim
    out = {
        "model.embed_tokens.weight": model.embed.cpu(),
        "model.norm.weight": model.final_norm_w.cpu(),
    }
    if not cfg.tie_embeddings:
        out["lm_head.weight"] = model.lm_head.cpu()
    for li, layer in enumerate(model.layers):
        p = f"model.layers.{li}."
        attn, mlp = layer["attn"], layer["mlp"]
        out[p + "input_layernorm.weight"] = layer["ln1"].cpu()
        out[p + "post_attention_layernorm.weight"] = layer["ln2"].cpu()
        if not hasattr(attn, "w_qkv"):  # GDN mixer
            lh, dk, dv = attn.lh, attn.dk, attn.dv
            w_in = attn.w_in.cpu()[:attn._proj_used]  # drop alignment pad
            out[p + "gdn.q_proj.weight"] = w_in[:lh * dk]
            out[p + "gdn.k_proj.weight"] = w_in[lh * dk:2 * lh * dk]
            out[p + "gdn.v_proj.weight"] = \
                w_in[2 * lh * dk:2 * lh * dk + lh * dv]
            out[p + "gdn.g_proj.weight"] = \
                w_in[2 * lh * dk + lh * dv:2 * lh * dk + lh * dv + lh]
            out[p + "gdn.b_proj.weight"] = w_in[2 * lh * dk + lh * dv + lh:]
            out[p + "gdn.o_proj.weight"] = attn.w_out.cpu()
            _save_mlp(out, p, layer["mlp"])
            continue
        qh, kvh = attn.qh, attn.kvh
        qkv = attn.w_qkv.cpu()
        out[p + "self_attn.q_proj.weight"] = qkv[:qh * d]
        out[p + "self_attn.k_proj.weight"] = qkv[qh * d:(qh + kvh) * d]
        out[p + "self_attn.v_proj.weight"] = qkv[(qh + kvh) * d:]
        out[p + "self_attn.o_proj.weight"] = attn.w_o.cpu()
        if cfg.qk_norm:
            out[p + "self_attn.q_norm.weight"] = attn.q_norm_w.cpu()
            out[p + "self_attn.k_norm.weight"] = attn.k_norm_w.cpu()
        if hasattr(mlp, "inter_shard"):
            _save_mlp(out, p, mlp)
        else:
            out[p + "mlp.gate.weight"] = mlp.router.cpu()
            for e in range(mlp.e_loc):
                ep = p + f"mlp.experts.{e}."
                gu = mlp.w_gate_up[e].cpu()
                inter = gu.shape[0] // 2
                out[ep + "gate_proj.weight"] = gu[:inter]
                out[ep + "up_proj.weight"] = gu[inter:]
                out[ep + "down_proj.weight"] = mlp.w_down[e].cpu()
    Path(path).mkdir(parents=True, exist_ok=True)
    save_file({k: v.contiguous() for k, v in out.items()},
              str(Path(path) / "model.safetensors"))
