"""GDN — gated delta net forward (Qwen3-Next linear-attention geometry).

Capability parity (behavior only) with Triton-distributed
python/triton_dist/kernels/nvidia/gdn.py:926-967 `chunk_gated_delta_rule_fwd`
(the FLA chunked gated-delta-rule: chunk_kkt_inv_ut -> state pass ->
output pass).

Recurrence (per head; state S in R^{K x V}):
    S_t   = exp(g_t) * S_{t-1}
    vhat  = beta_t * (v_t - S_t^T k_t)
    S_t  += k_t vhat^T
    o_t   = scale * S_t^T q_t

MI355X design: the prefill path is the chunked WY-form computed with
bf16 hipBLASLt matmuls + fp32 triangular solves (chunk 64) — on CDNA4
a hand-written kernel would re-derive the same GEMM chain, so the
library GEMMs ARE the fast path; the per-token decode step is a single
HIP kernel (csrc/kernels/gdn.hip) that keeps the [K,V] state resident
and is pure HBM-bound (state read+write).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch


def gated_delta_rule_recurrent_ref(
        q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
        g: torch.Tensor, beta: torch.Tensor, scale: float,
        initial_state: Optional[torch.Tensor] = None
) -> Tuple[torch.Tensor, torch.Tensor]:
    """fp32 token-recurrent golden reference.
    q,k: [B,T,H,K]; v: [B,T,H,V]; g (log decay): [B,T,H]; beta: [B,T,H].
    Returns (o [B,T,H,V], final_state [B,H,K,V])."""
    B, T, H, K = q.shape
    V = v.shape[-1]
    qf, kf, vf = q.float(), k.float(), v.float()
    gf, bf = g.float(), beta.float()
    S = (initial_state.float().clone() if initial_state is not None
         else torch.zeros(B, H, K, V))
    o = torch.zeros(B, T, H, V)
    for t in range(T):
        S = S * torch.exp(gf[:, t])[:, :, None, None]
        # S^T k : [B,H,V]
        sk = torch.einsum("bhkv,bhk->bhv", S, kf[:, t])
        vhat = bf[:, t][:, :, None] * (vf[:, t] - sk)
        S = S + torch.einsum("bhk,bhv->bhkv", kf[:, t], vhat)
        o[:, t] = scale * torch.einsum("bhkv,bhk->bhv", S, qf[:, t])
    return o, S


def chunk_gated_delta_rule_fwd(
        q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
        g: torch.Tensor, beta: torch.Tensor, scale: float,
        initial_state: Optional[torch.Tensor] = None,
        output_final_state: bool = True, chunk: int = 64
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Chunked WY-form forward (matches the recurrent reference).

    Per chunk with in-chunk log-decay cumsum gamma:
      A[i,j] = beta_i e^{gamma_i - gamma_j} (k_i . k_j)   (j < i)
      (I + A) vhat = beta*v - diag(beta e^gamma) K S0
      o_i  = scale * (e^{gamma_i} S0^T q_i
                      + sum_{j<=i} e^{gamma_i-gamma_j}(q_i.k_j) vhat_j)
      S'   = e^{gamma_C} S0 + K^T diag(e^{gamma_C-gamma}) vhat
    All contractions are plain batched GEMMs (hipBLASLt on GPU).
    """
    B, T, H, K = q.shape
    V = v.shape[-1]
    dev = q.device
    nc = (T + chunk - 1) // chunk
    pad = nc * chunk - T
    if pad:
        zp = lambda x: torch.cat(
            [x, torch.zeros(B, pad, *x.shape[2:], dtype=x.dtype,
                            device=dev)], dim=1)
        q, k, v, g, beta = zp(q), zp(k), zp(v), zp(g), zp(beta)
    # [B,nc,C,H,*] -> [B,H,nc,C,*]
    def chunked(x):
        return x.reshape(B, nc, chunk, H, *x.shape[3:]) \
                .permute(0, 3, 1, 2, *range(4, x.dim() + 1)).float()

    qc, kc, vc = chunked(q), chunked(k), chunked(v)
    gc, bc = chunked(g), chunked(beta)
    gam = gc.cumsum(-1)                       # [B,H,nc,C] inclusive cumsum
    lam = torch.exp(gam)
    lam_bc = bc * lam                         # beta_i e^{gamma_i}

    kk = torch.einsum("bhnik,bhnjk->bhnij", kc, kc)  # k_i . k_j
    ratio = torch.exp(gam[..., :, None] - gam[..., None, :])
    tril = torch.tril(torch.ones(chunk, chunk, device=dev), -1)
    A = bc[..., :, None] * ratio * kk * tril
    eye = torch.eye(chunk, device=dev)
    Tinv = torch.linalg.solve_triangular(A + eye, eye.expand_as(A),
                                         upper=False, unitriangular=True)

    S = (initial_state.float().clone() if initial_state is not None
         else torch.zeros(B, H, K, V, device=dev))
    o = torch.zeros(B, H, nc, chunk, V, device=dev)
    incl = torch.tril(torch.ones(chunk, chunk, device=dev))
    for c in range(nc):
        rhs = (bc[:, :, c, :, None] * vc[:, :, c]
               - lam_bc[:, :, c, :, None]
               * torch.einsum("bhik,bhkv->bhiv", kc[:, :, c], S))
        vhat = torch.einsum("bhij,bhjv->bhiv", Tinv[:, :, c], rhs)
        qk = torch.einsum("bhik,bhjk->bhij", qc[:, :, c], kc[:, :, c])
        m = ratio[:, :, c] * qk * incl
        o[:, :, c] = scale * (
            lam[:, :, c, :, None]
            * torch.einsum("bhik,bhkv->bhiv", qc[:, :, c], S)
            + torch.einsum("bhij,bhjv->bhiv", m, vhat))
        gc_end = gam[:, :, c, -1]
        decay = torch.exp(gc_end[..., None] - gam[:, :, c])  # [B,H,C]
        S = (torch.exp(gc_end)[..., None, None] * S
             + torch.einsum("bhik,bhi,bhiv->bhkv", kc[:, :, c], decay,
                            vhat))
    o = o.permute(0, 2, 3, 1, 4).reshape(B, nc * chunk, H, V)[:, :T]
    return o.to(v.dtype), (S if output_final_state else None)


def gdn_decode_step(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    g: torch.Tensor, beta: torch.Tensor, scale: float,
                    state: torch.Tensor) -> torch.Tensor:
    """One decode token: updates `state` [B,H,K,V] fp32 IN PLACE, returns
    o [B,H,V]. On GPU this is a single HIP kernel (block per (b,h),
    state-resident); on CPU the recurrent math directly."""
    if q.is_cuda:
        from .. import _C
        B, H, K = q.shape
        V = v.shape[-1]
        assert state.dtype == torch.float32 and state.is_contiguous()
        qb = q.to(torch.bfloat16).contiguous()
        kb = k.to(torch.bfloat16).contiguous()
        vb = v.to(torch.bfloat16).contiguous()
        gf = g.to(torch.float32).contiguous()
        bf = beta.to(torch.float32).contiguous()
        o = torch.empty(B, H, V, dtype=torch.bfloat16, device=q.device)
        _C.gdn_decode(qb.data_ptr(), kb.data_ptr(), vb.data_ptr(),
                      gf.data_ptr(), bf.data_ptr(), state.data_ptr(),
                      o.data_ptr(), B, H, K, V, scale,
                      torch.cuda.current_stream().cuda_stream)
        return o.to(q.dtype)
    qf, kf, vf = q.float(), k.float(), v.float()
    state.mul_(torch.exp(g.float())[:, :, None, None])
    sk = torch.einsum("bhkv,bhk->bhv", state, kf)
    vhat = beta.float()[:, :, None] * (vf - sk)
    state.add_(torch.einsum("bhk,bhv->bhkv", kf, vhat))
    return (scale * torch.einsum("bhkv,bhk->bhv", state, qf)).to(q.dtype)
