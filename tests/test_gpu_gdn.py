"""GPU GDN tests: the HIP decode-step kernel vs the fp32 recurrent
reference (Qwen3-Next geometry K=V=128), and the chunked prefill on GPU."""
import pytest
import torch
import torch.nn.functional as F

from triton_dist_amd.ops import (chunk_gated_delta_rule_fwd,
                                 gated_delta_rule_recurrent_ref,
                                 gdn_decode_step)

pytestmark = pytest.mark.gpu


def test_gdn_decode_kernel():
    B, T, H, K, V = 8, 6, 16, 128, 128
    g0 = torch.Generator().manual_seed(3)
    q = torch.randn(B, T, H, K, generator=g0)
    k = F.normalize(torch.randn(B, T, H, K, generator=g0), p=2, dim=-1)
    v = torch.randn(B, T, H, V, generator=g0)
    beta = torch.rand(B, T, H, generator=g0)
    g = F.logsigmoid(torch.rand(B, T, H, generator=g0))
    scale = K ** -0.5
    o_ref, s_ref = gated_delta_rule_recurrent_ref(q, k, v, g, beta, scale)

    state = torch.zeros(B, H, K, V, dtype=torch.float32, device="cuda")
    qd = q.to(torch.bfloat16).cuda()
    kd = k.to(torch.bfloat16).cuda()
    vd = v.to(torch.bfloat16).cuda()
    outs = []
    for t in range(T):
        outs.append(gdn_decode_step(qd[:, t], kd[:, t], vd[:, t],
                                    g[:, t].cuda(), beta[:, t].cuda(),
                                    scale, state))
    torch.cuda.synchronize()
    o = torch.stack(outs, 1).float().cpu()
    rel = (o - o_ref).abs().max() / o_ref.abs().max()
    assert rel < 0.05, rel
    srel = (state.cpu() - s_ref).abs().max() / s_ref.abs().max()
    assert srel < 0.05, srel


def test_gdn_chunked_gpu():
    B, T, H, K, V = 2, 200, 4, 128, 128
    g0 = torch.Generator().manual_seed(5)
    q = torch.randn(B, T, H, K, generator=g0)
    k = F.normalize(torch.randn(B, T, H, K, generator=g0), p=2, dim=-1)
    v = torch.randn(B, T, H, V, generator=g0)
    beta = torch.rand(B, T, H, generator=g0)
    g = F.logsigmoid(torch.rand(B, T, H, generator=g0))
    scale = K ** -0.5
    o_ref, s_ref = gated_delta_rule_recurrent_ref(q, k, v, g, beta, scale)
    o, s = chunk_gated_delta_rule_fwd(
        q.to(torch.bfloat16).cuda(), k.to(torch.bfloat16).cuda(),
        v.to(torch.bfloat16).cuda(), g.cuda(), beta.cuda(), scale)
    torch.cuda.synchronize()
    rel = (o.float().cpu() - o_ref).abs().max() / o_ref.abs().max()
    assert rel < 0.08, rel
