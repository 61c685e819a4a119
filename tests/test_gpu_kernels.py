"""Single-GPU kernel numerics + heap smoke tests (run on the MI355X box).

Every HIP kernel is compared against a plain PyTorch fp32 reference of the
same op.
"""
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
    _C.rs_reduce_bf16(segs.data_ptr(), out.data_ptr(), world, 3, m, n,
                      _stream())
    torch.cuda.synchronize()
    ref = segs.float().sum(0)
    assert_allclose(out, ref, atol=5e-1, rtol=2e-2)
