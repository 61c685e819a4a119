"""Single-GPU kernel numerics + heap smoke tests (run on the MI355X box).

Every HIP kernel is compared against a plain PyTorch fp32 reference of the
same op.
"""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def heap():
    import triton_dist_amd as td

    td.initialize_distributed()
    h = td.init_symm_heap()
    yield h
    td.shutdown_heap()


def _stream():
    return torch.cuda.current_stream().cuda_stream


def test_mfma_layout_probe():
    """Determine the gfx950 mfma_f32_16x16x32_bf16 operand k-mapping and
    assert the compiled-in TD_MFMA_KLANE (=0) is the correct one."""
    from triton_dist_amd import _C

    torch.manual_seed(0)
    a = torch.randn(16, 32, device="cuda").to(torch.bfloat16).contiguous()
    b = torch.randn(32, 16, device="cuda").to(torch.bfloat16).contiguous()
    ref = a.float() @ b.float()
    good = []
    for layout in (0, 1):
        c = torch.zeros(16, 16, device="cuda", dtype=torch.float32)
        _C.probe_mfma(a.data_ptr(), b.data_ptr(), c.data_ptr(), layout,
                      _stream())
        torch.cuda.synchronize()
        if torch.allclose(c, ref, atol=1e-1, rtol=1e-2):
            good.append(layout)
    assert good, "neither MFMA operand layout candidate matched the reference"
    assert 0 in good, (
        f"TD_MFMA_KLANE must be set to {good[0]} (layout 0 did not match)")


def test_heap_views_world1(heap):
    buf = heap.alloc_buffer((128, 64), torch.bfloat16)
    t = buf.local()
    assert t.is_cuda and t.dtype == torch.bfloat16
    src = torch.randn(128, 64, device="cuda").to(torch.bfloat16)
    t.copy_(src)
    torch.cuda.synchronize()
    assert torch.equal(buf.local(), src)


def test_barrier_world1(heap):
    for _ in range(5):
        heap.barrier_all_on_stream()
    torch.cuda.synchronize()


def test_put_signal_wait(heap):
    from triton_dist_amd import _C

    n = 1 << 20
    src = heap.alloc_buffer((n,), torch.uint8)
    dst = heap.alloc_buffer((n,), torch.uint8)
    flag = heap.alloc_buffer((4,), torch.int32)
    src.local().copy_(torch.arange(n, device="cuda", dtype=torch.int32)
                      .remainder(251).to(torch.uint8))
    torch.cuda.synchronize()
    s = _stream()
    _C.put_signal(dst.ptr(), src.ptr(), n, flag.ptr(), 7, False, s)
    _C.wait_eq(flag.ptr(), 1, 7, s)
    torch.cuda.synchronize()
    assert torch.equal(dst.local(), src.local())
    assert int(flag.local()[0]) == 7


def test_copy_kernel(heap):
    from triton_dist_amd import _C

    x = torch.randn(3333, 257, device="cuda")
    y = torch.empty_like(x)
    _C.copy_kernel(y.data_ptr(), x.data_ptr(), x.numel() * 4, _stream())
    torch.cuda.synchronize()
    assert torch.equal(x, y)


@pytest.mark.parametrize("m,n,k", [(128, 128, 64), (256, 384, 128),
                                   (1024, 1024, 512), (4096, 3584, 5120)])
def test_gemm_bf16(m, n, k):
    from triton_dist_amd.ops import gemm
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    torch.manual_seed(m + n + k)
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16) / 8
    w = torch.randn(n, k, device="cuda").to(torch.bfloat16) / 8
    c = gemm(a, w)
    ref = a.float() @ w.float().t()
    assert_allclose(c, ref, **bf16_gemm_tol(k))


def test_gemm_bias():
    from triton_dist_amd.ops import gemm
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    m, n, k = 256, 256, 256
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16) / 8
    w = torch.randn(n, k, device="cuda").to(torch.bfloat16) / 8
    bias = torch.randn(n, device="cuda").to(torch.bfloat16)
    c = gemm(a, w, bias=bias)
    ref = a.float() @ w.float().t() + bias.float()
    assert_allclose(c, ref, **bf16_gemm_tol(k))


def test_rs_reduce():
    from triton_dist_amd import _C
    from triton_dist_amd.utils.testing import assert_allclose

    world, m, n = 8, 256, 512
    segs = torch.randn(world, m, n, device="cuda").to(torch.bfloat16)
    out = torch.empty(m, n, device="cuda", dtype=torch.bfloat16)
    _C.rs_reduce_bf16(segs.data_ptr(), out.data_ptr(), world, 3, m, m, n,
                      _stream())
    torch.cuda.synchronize()
    ref = segs.float().sum(0)
    assert_allclose(out, ref, atol=5e-1, rtol=2e-2)


def test_rmsnorm_kernels():
    from triton_dist_amd.ops.fused import add_rms_norm_op, rms_norm_op
    from triton_dist_amd.utils.testing import assert_allclose

    torch.manual_seed(0)
    x = torch.randn(333, 5120, device="cuda").to(torch.bfloat16)
    w = torch.randn(5120, device="cuda").to(torch.bfloat16)
    out = rms_norm_op(x, w)
    x32 = x.float()
    ref = x32 * torch.rsqrt(x32.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()
    assert_allclose(out, ref, atol=3e-2, rtol=3e-2)

    d = torch.randn_like(x)
    nr, nm = add_rms_norm_op(d, x, w)
    y = (x.float() + d.float())
    assert_allclose(nr, y.to(torch.bfloat16), atol=2e-2, rtol=2e-2)
    ref2 = y * torch.rsqrt(y.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()
    assert_allclose(nm, ref2, atol=3e-2, rtol=3e-2)


def test_swiglu_kernel():
    import torch.nn.functional as F

    from triton_dist_amd.ops.fused import swiglu_op
    from triton_dist_amd.utils.testing import assert_allclose

    torch.manual_seed(1)
    h = torch.randn(256, 2 * 3200, device="cuda").to(torch.bfloat16)
    out = swiglu_op(h, 3200)
    ref = F.silu(h[:, :3200].float()) * h[:, 3200:].float()
    assert_allclose(out, ref, atol=3e-2, rtol=3e-2)


def test_qkv_prologue_and_flash_decode():
    import torch.nn.functional as F

    from triton_dist_amd.layers.norm import Rotary, rms_norm
    from triton_dist_amd.ops.fused import (flash_decode_op,
                                           qkv_prologue_decode_op)
    from triton_dist_amd.utils.testing import assert_allclose

    torch.manual_seed(2)
    b, qh, kvh, d, maxlen = 16, 8, 1, 128, 96
    seq = 37  # existing cache length; new token at position 37
    rot = Rotary(d, maxlen, device="cuda")
    qnw = torch.randn(d, device="cuda").abs().to(torch.bfloat16)
    knw = torch.randn(d, device="cuda").abs().to(torch.bfloat16)
    qkv = (torch.randn(b, (qh + 2 * kvh) * d, device="cuda") / 4).to(torch.bfloat16)
    kc = (torch.randn(b, maxlen, kvh, d, device="cuda") / 4).to(torch.bfloat16)
    vc = (torch.randn(b, maxlen, kvh, d, device="cuda") / 4).to(torch.bfloat16)
    kc_ref, vc_ref = kc.clone(), vc.clone()
    offset = torch.tensor(seq, dtype=torch.int64, device="cuda")

    q_rot = qkv_prologue_decode_op(qkv, kc, vc, offset, rot.cos, rot.sin,
                                   qnw, knw, qh, kvh, 1e-6, True)
    torch.cuda.synchronize()

    # torch reference of the prologue
    q = qkv[:, :qh * d].view(b, 1, qh, d)
    k = qkv[:, qh * d:(qh + kvh) * d].view(b, 1, kvh, d)
    v = qkv[:, (qh + kvh) * d:].view(b, 1, kvh, d)
    qn = rms_norm(q, qnw)
    kn = rms_norm(k, knw)
    pos = torch.full((b, 1), seq, device="cuda", dtype=torch.int64)
    qr, kr = rot.apply(qn, kn, pos)
    assert_allclose(q_rot.view(b, 1, qh, d), qr, atol=4e-2, rtol=4e-2)
    assert_allclose(kc[:, seq], kr[:, 0], atol=4e-2, rtol=4e-2)
    assert_allclose(vc[:, seq], v[:, 0], atol=1e-3, rtol=1e-3)
    # untouched cache rows stay untouched
    assert torch.equal(kc[:, :seq], kc_ref[:, :seq])
    assert torch.equal(vc[:, seq + 1:], vc_ref[:, seq + 1:])

    # flash decode vs sdpa over the valid prefix
    out = flash_decode_op(q_rot, kc, vc, offset, qh, kvh)
    torch.cuda.synchronize()
    qs = q_rot.view(b, qh, 1, d).float()
    ks = kc[:, :seq + 1].transpose(1, 2).float()  # [b, kvh, L, d]
    vs = vc[:, :seq + 1].transpose(1, 2).float()
    ref = F.scaled_dot_product_attention(qs, ks, vs, enable_gqa=True)
    assert_allclose(out.view(b, qh, 1, d), ref, atol=4e-2, rtol=4e-2)


@pytest.mark.parametrize("m,n,k", [(8192, 11008, 4096),
                                   (8192, 4096, 12288)])
def test_gemm_ci_shapes(m, n, k):
    """The reference's own CI regression shapes (amd-ci.yml:
    test_ag_gemm_intra_node 8192 11008 4096 / test_gemm_rs_intra_node
    8192 4096 12288) as plain-GEMM numerics checks — the fused 2-rank
    forms run the same consumer kernels with the same tile math (and are
    covered at smaller shapes by test_gpu_dist; running THESE grids
    2-ranks-on-1-GPU would oversubscribe the dev box)."""
    from triton_dist_amd.ops import gemm
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    torch.manual_seed(n)
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16) / 8
    w = torch.randn(n, k, device="cuda").to(torch.bfloat16) / 8
    c = gemm(a, w)
    ref = a.float() @ w.float().t()
    assert_allclose(c, ref, **bf16_gemm_tol(k))


@pytest.mark.parametrize("qh,kvh", [(10, 1), (16, 1), (12, 2)])
def test_flash_decode_wide_group(qh, kvh):
    """G = qh/kvh in (10, 16, 6): the 2-pass softmax sweep (seed-oss-36b
    TP1 geometry has G=10)."""
    import torch.nn.functional as F

    from triton_dist_amd.ops.fused import flash_decode_op
    from triton_dist_amd.utils.testing import assert_allclose

    torch.manual_seed(qh * 31 + kvh)
    b, d, maxlen, seq = 8, 128, 80, 51
    q = (torch.randn(b, qh * d, device="cuda") / 4).to(torch.bfloat16)
    kc = (torch.randn(b, maxlen, kvh, d, device="cuda") / 4).to(
        torch.bfloat16)
    vc = (torch.randn(b, maxlen, kvh, d, device="cuda") / 4).to(
        torch.bfloat16)
    offset = torch.tensor(seq - 1, dtype=torch.int64, device="cuda")
    out = flash_decode_op(q, kc, vc, offset, qh, kvh)
    torch.cuda.synchronize()
    qs = q.view(b, qh, 1, d).float()
    ks = kc[:, :seq].transpose(1, 2).float()
    vs = vc[:, :seq].transpose(1, 2).float()
    ref = F.scaled_dot_product_attention(qs, ks, vs, enable_gqa=True)
    assert_allclose(out.view(b, qh, 1, d), ref, atol=4e-2, rtol=4e-2)


@pytest.mark.parametrize("m,n,k", [(512, 1280, 5120), (512, 6400, 5120),
                                   (256, 5120, 1024), (128, 128, 4096)])
def test_gemm_splitk(m, n, k):
    from triton_dist_amd.ops import gemm
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    torch.manual_seed(m + n + k)
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16) / 8
    w = torch.randn(n, k, device="cuda").to(torch.bfloat16) / 8
    # shapes route to the skinny or split-K occupancy tiers
    c = gemm(a, w)
    ref = a.float() @ w.float().t()
    assert_allclose(c, ref, **bf16_gemm_tol(k))


@pytest.mark.skipif(not os.environ.get("TD_EXPERIMENTAL"),
                    reason="experimental kernel: set TD_EXPERIMENTAL=1")
@pytest.mark.parametrize("m,n,k", [(256, 256, 128), (512, 768, 512),
                                   (4096, 4096, 4096)])
def test_gemm256_v2_experimental(m, n, k):
    """BK=64 quadrant-phase template (gemm256_v2.hip) — round-2 WIP."""
    from triton_dist_amd import _C
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    torch.manual_seed(k)
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16) / 8
    w = torch.randn(n, k, device="cuda").to(torch.bfloat16) / 8
    c = torch.empty(m, n, device="cuda", dtype=torch.bfloat16)
    _C.gemm256_v2_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(), m, n, k,
                       torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    assert_allclose(c, a.float() @ w.float().t(), **bf16_gemm_tol(k))


@pytest.mark.parametrize("m,n,k", [(256, 256, 128), (512, 768, 512),
                                   (512, 768, 640), (4096, 4096, 4096)])
def test_gemm256_v3(m, n, k):
    """Faithful 8-phase template rebuild (gemm256_v3.hip): register-reuse
    gray quadrant walk + counted vmcnt drains. Ledger CPU-proven in
    test_mappings_cpu.test_gemm256_v3_pipeline_ledger."""
    from triton_dist_amd import _C
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    torch.manual_seed(k)
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16) / 8
    w = torch.randn(n, k, device="cuda").to(torch.bfloat16) / 8
    c = torch.empty(m, n, device="cuda", dtype=torch.bfloat16)
    for _ in range(3):  # re-run: race screen, not just one lucky pass
        c.zero_()
        _C.gemm256_v3_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(), m, n,
                           k, torch.cuda.current_stream().cuda_stream)
        torch.cuda.synchronize()
        assert_allclose(c, a.float() @ w.float().t(), **bf16_gemm_tol(k))


@pytest.mark.skipif(not os.environ.get("TD_EXPERIMENTAL"),
                    reason="experimental kernel: set TD_EXPERIMENTAL=1")
def test_flash_decode_paged_experimental():
    """Paged pool + block table vs the contiguous kernel."""
    from triton_dist_amd.models.kv_cache import PagedKVCache
    from triton_dist_amd.ops.fused import (flash_decode_op,
                                           flash_decode_paged_op)
    from triton_dist_amd.utils.testing import assert_allclose

    torch.manual_seed(9)
    b, qh, kvh, d, maxlen, seq = 8, 16, 2, 128, 256, 77
    paged = PagedKVCache(1, b, maxlen, kvh, d, block=64, device="cuda")
    kc = (torch.randn(b, maxlen, kvh, d, device="cuda") / 4).to(
        torch.bfloat16)
    vc = (torch.randn(b, maxlen, kvh, d, device="cuda") / 4).to(
        torch.bfloat16)
    paged.append(0, kc[:, :seq], vc[:, :seq], 0)
    q = (torch.randn(b, qh * d, device="cuda") / 4).to(torch.bfloat16)
    offset = torch.tensor(seq - 1, dtype=torch.int64, device="cuda")
    out_p = flash_decode_paged_op(q, paged, 0, offset, qh, kvh)
    out_c = flash_decode_op(q, kc, vc, offset, qh, kvh)
    torch.cuda.synchronize()
    assert_allclose(out_p, out_c, atol=1e-3, rtol=1e-3)


@pytest.mark.parametrize("b,s,qh,kvh,causal", [
    (2, 128, 8, 2, True),
    (2, 128, 8, 2, False),
    (1, 256, 4, 4, True),     # two q-tiles, MHA
    (2, 100, 8, 2, True),     # ragged s: row/col masking
    (1, 128, 64, 8, True),    # qwen3-32b TP1 head geometry
    (2, 37, 10, 2, True),     # G=5, s smaller than a tile
])
def test_flash_prefill(b, s, qh, kvh, causal):
    """MFMA FA2 prefill kernel vs fp32 sdpa (attention.hip
    k_flash_prefill)."""
    import torch.nn.functional as F
    from triton_dist_amd.ops.fused import flash_prefill_op
    from triton_dist_amd.utils.testing import assert_allclose

    torch.manual_seed(s * 31 + qh)
    d = 128
    q = (torch.randn(b, s, qh, d, device="cuda") / 4).to(torch.bfloat16)
    k = (torch.randn(b, s, kvh, d, device="cuda") / 4).to(torch.bfloat16)
    v = (torch.randn(b, s, kvh, d, device="cuda") / 4).to(torch.bfloat16)
    out, lse = flash_prefill_op(q, k, v, causal=causal, return_lse=True)
    torch.cuda.synchronize()
    ref = F.scaled_dot_product_attention(
        q.permute(0, 2, 1, 3).float(), k.permute(0, 2, 1, 3).float(),
        v.permute(0, 2, 1, 3).float(), is_causal=causal, enable_gqa=True
    ).permute(0, 2, 1, 3)
    assert_allclose(out, ref, atol=3e-2, rtol=3e-2)
    # LSE check vs manual fp32 computation
    scale = 1.0 / d ** 0.5
    G = qh // kvh
    kk = k.repeat_interleave(G, dim=2).permute(0, 2, 1, 3).float()
    sc = torch.einsum("bhqd,bhkd->bhqk",
                      q.permute(0, 2, 1, 3).float(), kk) * scale
    if causal:
        mask = torch.arange(s, device="cuda")
        sc = sc.masked_fill(mask[None, None, None, :]
                            > mask[None, None, :, None], float("-inf"))
    ref_lse = torch.logsumexp(sc, dim=-1).permute(0, 2, 1)  # [b,s,qh]
    assert_allclose(lse, ref_lse, atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("m,n,k,fuse", [
    (512, 1280, 5120, 0),
    (32, 128, 64, 0),
    (512, 512, 27648, 0),
    (256, 1024, 2048, 1),    # fused swiglu: interleaved gate/up pairs
])
def test_gemm_skinny(m, n, k, fuse):
    """Skinny-M decode tier (gemm_skinny.hip) vs fp32 reference, incl.
    the fused-SwiGLU epilogue on interleaved columns."""
    import torch.nn.functional as F
    from triton_dist_amd import _C
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    torch.manual_seed(m + n)
    a = (torch.randn(m, k, device="cuda") / 8).to(torch.bfloat16)
    w = (torch.randn(n, k, device="cuda") / 8).to(torch.bfloat16)
    ncols = n // 2 if fuse else n
    c = torch.empty(m, ncols, device="cuda", dtype=torch.bfloat16)
    _C.gemm_skinny_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(), 0,
                        m, n, k, fuse, torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    full = a.float() @ w.float().t()
    if fuse:
        ref = F.silu(full[:, 0::2]) * full[:, 1::2]
    else:
        ref = full
    assert_allclose(c, ref, **bf16_gemm_tol(k))


def test_gemm256_sk2():
    """Two-stage split-K tier (private fp32 slices + fused reduce) vs
    fp32 reference and the atomic tier."""
    from triton_dist_amd import _C
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    torch.manual_seed(9)
    m, n, k = 512, 1280, 5120
    a = (torch.randn(m, k, device="cuda") / 8).to(torch.bfloat16)
    w = (torch.randn(n, k, device="cuda") / 8).to(torch.bfloat16)
    s = torch.cuda.current_stream().cuda_stream
    for sk in (2, 4, 8):
        ws = torch.empty(sk, m, n, dtype=torch.float32, device="cuda")
        c = torch.empty(m, n, device="cuda", dtype=torch.bfloat16)
        _C.gemm256_sk2_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(), 0,
                            ws.data_ptr(), m, n, k, sk, s)
        torch.cuda.synchronize()
        assert_allclose(c, a.float() @ w.float().t(), msg=f"sk={sk}",
                        **bf16_gemm_tol(k))


def test_gemm_stream():
    """Weight-streaming decode GEMM (BM=512 register-K tier; kept as a
    measured-negative experiment, profiles/README.md) vs fp32 ref:
    full and partial m, bias, and the split-K reduce path."""
    from triton_dist_amd import _C
    from triton_dist_amd.utils.testing import assert_allclose, bf16_gemm_tol

    torch.manual_seed(11)
    s = torch.cuda.current_stream().cuda_stream
    for m, n, k, sk, use_bias in [(512, 1536, 2048, 1, False),
                                  (512, 1536, 2048, 4, False),
                                  (300, 640, 1024, 1, True),
                                  (300, 640, 1024, 2, True)]:
        a = (torch.randn(m, k, device="cuda") / 8).to(torch.bfloat16)
        w = (torch.randn(n, k, device="cuda") / 8).to(torch.bfloat16)
        bias = (torch.randn(n, device="cuda") / 4).to(torch.bfloat16) \
            if use_bias else None
        c = torch.empty(m, n, device="cuda", dtype=torch.bfloat16)
        ws = torch.empty(sk, m, n, dtype=torch.float32, device="cuda") \
            if sk > 1 else c
        _C.gemm_stream_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(),
                            bias.data_ptr() if use_bias else 0,
                            ws.data_ptr(), m, n, k, sk, s)
        torch.cuda.synchronize()
        ref = a.float() @ w.float().t()
        if use_bias:
            ref = ref + bias.float()
        assert_allclose(c, ref, msg=f"m={m} sk={sk}", **bf16_gemm_tol(k))


def test_moe_pq_nbuf_variants():
    """k_moe_grouped_gemm_pq<3>/<4> (env-gated A/B variants) match <2>
    numerics on a grouped decode shape."""
    import subprocess
    import sys
    code = r'''
import torch
from triton_dist_amd import _C
torch.manual_seed(5)
E, rows_e, n, k = 8, 32, 256, 512
rows = E * rows_e
x = (torch.randn(rows + 128, k, device="cuda") / 8).to(torch.bfloat16)
w = (torch.randn(E, n, k, device="cuda") / 8).to(torch.bfloat16)
base = torch.arange(E, dtype=torch.int32, device="cuda") * rows_e
erows = torch.full((E,), rows_e, dtype=torch.int32, device="cuda")
items = torch.tensor([(e << 16) for e in range(E)], dtype=torch.int32,
                     device="cuda")
cnt = torch.tensor([E], dtype=torch.int32, device="cuda")
out = torch.zeros(rows + 128, n, dtype=torch.bfloat16, device="cuda")
s = torch.cuda.current_stream().cuda_stream
_C.moe_grouped_gemm_pq(x.data_ptr(), w.data_ptr(), out.data_ptr(),
                       base.data_ptr(), erows.data_ptr(),
                       items.data_ptr(), cnt.data_ptr(), n, k, s)
torch.cuda.synchronize()
ref = torch.cat([x[e * rows_e:(e + 1) * rows_e].float()
                 @ w[e].float().t() for e in range(E)])
err = (out[:rows].float() - ref).abs().max().item()
assert err < 0.1, err
print("OK")
'''
    import os as _os
    for env in ({}, {"TD_MOE_PQ3": "1"}, {"TD_MOE_PQ4": "1"}):
        r = subprocess.run([sys.executable, "-c", code],
                           env={**_os.environ, **env},
                           capture_output=True, text=True, timeout=180)
        assert r.returncode == 0 and "OK" in r.stdout, \
            (env, r.stdout[-500:], r.stderr[-500:])
