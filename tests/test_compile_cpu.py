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
