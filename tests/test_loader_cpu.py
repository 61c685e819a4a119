"""Round-trip + shard tests for the HF-checkpoint loader: save a
random-init model in HF naming, reload it (world=1 equality; world=2
shard correctness vs manual slices)."""
import tempfile

import torch

from tests.conftest import run_distributed


def _worker_roundtrip(rank, world, tmpdir):
    import triton_dist_amd as td
    from triton_dist_amd.models import (AutoLLM, get_config,
                                        load_hf_weights, save_hf_weights)

    td.init_symm_heap(size_mb=16)
    cfg = get_config("tiny", tp_mode="torch")
    if world == 1:
        a = AutoLLM(cfg, device="cpu")
        a.init_weights(seed=5)
        save_hf_weights(a, tmpdir)
        b = AutoLLM(cfg, device="cpu")
        n = load_hf_weights(b, tmpdir)
        assert n > 0
        assert torch.equal(a.embed, b.embed)
        for la, lb in zip(a.layers, b.layers):
            assert torch.equal(la["attn"].w_qkv, lb["attn"].w_qkv)
            assert torch.equal(la["mlp"].w_gate_up, lb["mlp"].w_gate_up)
            assert torch.equal(la["mlp"].w_down, lb["mlp"].w_down)
            assert torch.equal(la["ln1"], lb["ln1"])
    else:
        # every rank loads the same file; check shards match manual slices
        m = AutoLLM(cfg, device="cpu")
        load_hf_weights(m, tmpdir)
        from safetensors import safe_open
        with safe_open(f"{tmpdir}/model.safetensors", framework="pt") as sf:
            wq = sf.get_tensor("model.layers.0.self_attn.q_proj.weight")
            wd = sf.get_tensor("model.layers.0.mlp.down_proj.weight")
        attn = m.layers[0]["attn"]
        d = cfg.head_dim
        exp_q = wq[rank * attn.qh * d:(rank + 1) * attn.qh * d]
        assert torch.equal(attn.w_qkv[:attn.qh * d].float(), exp_q.float())
        mlp = m.layers[0]["mlp"]
        i_s = mlp.inter_shard
        exp_d = wd[:, rank * i_s:(rank + 1) * i_s]
        assert torch.equal(mlp.w_down.float(), exp_d.float())
    td.shutdown_heap()


def test_loader_roundtrip_and_shard():
    with tempfile.TemporaryDirectory() as tmp:
        run_distributed(_worker_roundtrip, world_size=1, args=(tmp,))
        run_distributed(_worker_roundtrip, world_size=2, args=(tmp,))


def _worker_moe(rank, world, tmpdir):
    import triton_dist_amd as td
    from triton_dist_amd.models import (AutoLLM, get_config,
                                        load_hf_weights, save_hf_weights)

    td.init_symm_heap(size_mb=16)
    cfg = get_config("tiny-moe", tp_mode="torch")
    if world == 1:
        a = AutoLLM(cfg, device="cpu")
        a.init_weights(seed=9)
        save_hf_weights(a, tmpdir)
        b = AutoLLM(cfg, device="cpu")
        load_hf_weights(b, tmpdir)
        for la, lb in zip(a.layers, b.layers):
            assert torch.equal(la["mlp"].router, lb["mlp"].router)
            assert torch.equal(la["mlp"].w_gate_up, lb["mlp"].w_gate_up)
    else:
        m = AutoLLM(cfg, device="cpu")
        load_hf_weights(m, tmpdir)
        from safetensors import safe_open
        lo = rank * m.layers[0]["mlp"].e_loc
        with safe_open(f"{tmpdir}/model.safetensors", framework="pt") as sf:
            g0 = sf.get_tensor(
                f"model.layers.0.mlp.experts.{lo}.gate_proj.weight")
        mlp = m.layers[0]["mlp"]
        inter = mlp.w_gate_up.shape[1] // 2
        assert torch.equal(mlp.w_gate_up[0, :inter].float(), g0.float())
    td.shutdown_heap()


def test_loader_moe():
    with tempfile.TemporaryDirectory() as tmp:
        run_distributed(_worker_moe, world_size=1, args=(tmp,))
        run_distributed(_worker_moe, world_size=2, args=(tmp,))


def _worker_index(rank, world, tmpdir):
    """Sharded-checkpoint form: weight_map index over two files."""
    import json
    from pathlib import Path

    import torch
    import triton_dist_amd as td
    from triton_dist_amd.models import (AutoLLM, get_config,
                                        load_hf_weights, save_hf_weights)

    td.init_symm_heap(size_mb=16)
    cfg = get_config("tiny", tp_mode="torch")
    a = AutoLLM(cfg, device="cpu")
    a.init_weights(seed=11)
    save_hf_weights(a, tmpdir)
    # split the single file into two + an index
    from safetensors import safe_open
    from safetensors.torch import save_file
    src = Path(tmpdir) / "model.safetensors"
    with safe_open(str(src), framework="pt") as sf:
        keys = sorted(sf.keys())
        half = len(keys) // 2
        t1 = {k: sf.get_tensor(k) for k in keys[:half]}
        t2 = {k: sf.get_tensor(k) for k in keys[half:]}
    src.unlink()
    save_file(t1, str(Path(tmpdir) / "model-00001-of-00002.safetensors"))
    save_file(t2, str(Path(tmpdir) / "model-00002-of-00002.safetensors"))
    wm = {k: "model-00001-of-00002.safetensors" for k in keys[:half]}
    wm.update({k: "model-00002-of-00002.safetensors" for k in keys[half:]})
    (Path(tmpdir) / "model.safetensors.index.json").write_text(
        json.dumps({"weight_map": wm}))

    b = AutoLLM(cfg, device="cpu")
    load_hf_weights(b, tmpdir)
    assert torch.equal(a.embed, b.embed)
    assert torch.equal(a.layers[1]["mlp"].w_down, b.layers[1]["mlp"].w_down)
    td.shutdown_heap()


def test_loader_sharded_index():
    import tempfile
    with tempfile.TemporaryDirectory() as tmp:
        run_distributed(_worker_index, world_size=1, args=(tmp,))


def _worker_hybrid(rank, world, tmpdir):
    import torch
    import triton_dist_amd as td
    from triton_dist_amd.models import (AutoLLM, get_config,
                                        load_hf_weights, save_hf_weights)

    td.init_symm_heap(size_mb=16)
    cfg = get_config("tiny-gdn", tp_mode="torch")
    if world == 1:
        a = AutoLLM(cfg, device="cpu")
        a.init_weights(seed=21)
        save_hf_weights(a, tmpdir)
        b = AutoLLM(cfg, device="cpu")
        load_hf_weights(b, tmpdir)
        for la, lb in zip(a.layers, b.layers):
            aa, ab = la["attn"], lb["attn"]
            if hasattr(aa, "w_in"):
                assert torch.equal(aa.w_in, ab.w_in)
                assert torch.equal(aa.w_out, ab.w_out)
            else:
                assert torch.equal(aa.w_qkv, ab.w_qkv)
    else:
        m = AutoLLM(cfg, device="cpu")
        load_hf_weights(m, tmpdir)  # sharded load of the world-1 save
    td.shutdown_heap()


def test_loader_hybrid_gdn():
    import tempfile
    with tempfile.TemporaryDirectory() as tmp:
        run_distributed(_worker_hybrid, world_size=1, args=(tmp,))
        run_distributed(_worker_hybrid, world_size=2, args=(tmp,))
