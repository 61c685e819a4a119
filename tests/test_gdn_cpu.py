"""GDN (gated delta net) CPU tests: chunked WY-form forward vs the
token-recurrent golden reference; decode-step chain equivalence."""
import torch
import torch.nn.functional as F

from triton_dist_amd.ops import (chunk_gated_delta_rule_fwd,
                                 gated_delta_rule_recurrent_ref,
                                 gdn_decode_step)


def _inputs(B=2, T=130, H=3, K=32, V=16, seed=0):
    g0 = torch.Generator().manual_seed(seed)
    q = torch.randn(B, T, H, K, generator=g0)
    k = F.normalize(torch.randn(B, T, H, K, generator=g0), p=2, dim=-1)
    v = torch.randn(B, T, H, V, generator=g0)
    beta = torch.rand(B, T, H, generator=g0)
    g = F.logsigmoid(torch.rand(B, T, H, generator=g0))
    return q, k, v, g, beta, K ** -0.5


def test_chunked_matches_recurrent():
    q, k, v, g, beta, scale = _inputs()
    o_ref, s_ref = gated_delta_rule_recurrent_ref(q, k, v, g, beta, scale)
    o, s = chunk_gated_delta_rule_fwd(q, k, v, g, beta, scale, chunk=64)
    assert (o.float() - o_ref).abs().max() < 1e-4
    assert (s - s_ref).abs().max() < 1e-4


def test_chunked_with_initial_state():
    q, k, v, g, beta, scale = _inputs(T=65)
    s0 = torch.randn(2, 3, 32, 16)
    o_ref, s_ref = gated_delta_rule_recurrent_ref(q, k, v, g, beta, scale,
                                                  initial_state=s0)
    o, s = chunk_gated_delta_rule_fwd(q, k, v, g, beta, scale,
                                      initial_state=s0, chunk=64)
    assert (o.float() - o_ref).abs().max() < 1e-4
    assert (s - s_ref).abs().max() < 1e-4


def test_decode_chain_equals_recurrent():
    q, k, v, g, beta, scale = _inputs(T=12)
    o_ref, s_ref = gated_delta_rule_recurrent_ref(q, k, v, g, beta, scale)
    state = torch.zeros(2, 3, 32, 16)
    outs = [gdn_decode_step(q[:, t], k[:, t], v[:, t], g[:, t],
                            beta[:, t], scale, state)
            for t in range(12)]
    o = torch.stack(outs, 1)
    assert (o.float() - o_ref).abs().max() < 1e-5
    assert (state - s_ref).abs().max() < 1e-5
