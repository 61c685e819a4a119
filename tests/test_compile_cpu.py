"""torch.compile integration: custom ops trace through Dynamo (CPU uses
the torch fallback impls)."""
import torch


def test_custom_ops_compile():
    from triton_dist_amd.compile import register_custom_ops

    register_custom_ops()

    def fn(x, w):
        y = torch.ops.td.rms_norm(x, w, 1e-6)
        h = torch.cat([y, y * 2], dim=-1)
        return torch.ops.td.swiglu(h, x.shape[-1])

    x = torch.randn(8, 64).to(torch.bfloat16)
    w = torch.ones(64).to(torch.bfloat16)
    eager = fn(x, w)
    compiled = torch.compile(fn, backend="eager")(x, w)
    assert torch.equal(eager, compiled)


def test_custom_ops_gdn_add_rms():
    import torch.nn.functional as F

    from triton_dist_amd.compile import register_custom_ops
    from triton_dist_amd.ops import gated_delta_rule_recurrent_ref

    register_custom_ops()

    # add_rms_norm through compile
    def fn(x, r, w):
        nr, y = torch.ops.td.add_rms_norm(x, r, w, 1e-6)
        return nr + y

    x = torch.randn(4, 32).to(torch.bfloat16)
    r = torch.randn(4, 32).to(torch.bfloat16)
    w = torch.ones(32).to(torch.bfloat16)
    assert torch.equal(fn(x, r, w),
                       torch.compile(fn, backend="eager")(x, r, w))

    # gdn_decode (state-mutating custom op) matches the recurrent ref
    B, H, K, V = 2, 3, 16, 8
    g0 = torch.Generator().manual_seed(1)
    q = torch.randn(B, 1, H, K, generator=g0)
    k = F.normalize(torch.randn(B, 1, H, K, generator=g0), p=2, dim=-1)
    v = torch.randn(B, 1, H, V, generator=g0)
    beta = torch.rand(B, 1, H, generator=g0)
    g = F.logsigmoid(torch.rand(B, 1, H, generator=g0))
    o_ref, s_ref = gated_delta_rule_recurrent_ref(q, k, v, g, beta, 0.25)
    state = torch.zeros(B, H, K, V)
    o = torch.ops.td.gdn_decode(q[:, 0], k[:, 0], v[:, 0], g[:, 0],
                                beta[:, 0], 0.25, state)
    assert (o.float() - o_ref[:, 0]).abs().max() < 1e-5
    assert (state - s_ref).abs().max() < 1e-5
