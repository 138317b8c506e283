"""Gated DeltaNet (GDN) linear-attention reference ops.

Parity target: the reference's vendored FLA Triton kernels
(layers/ops/fla/: fused_gdn_gating, fused_recurrent_gated_delta_rule,
chunk_gated_delta_rule) and causal_conv1d (layers/ops/mamba/). These
torch implementations are the round-1 compute path (CPU and GPU) and
the numerics oracle for the round-2 HIP chunked kernels.

Recurrence per v-head (state S [Dv, Dk]; q/k L2-normalized, q scaled):
    S  = S * exp(g_t)
    u  = beta_t * (v_t - S @ k_t)
    S  = S + outer(u, k_t)
    o_t = S @ q_t
GVA: v-heads group over k-heads (Hv = G * Hk); gating g/beta are
per-v-head.
"""

from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def gdn_gating(A_log: torch.Tensor, a: torch.Tensor, b: torch.Tensor,
               dt_bias: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """g = -exp(A_log) * softplus(a + dt_bias); beta = sigmoid(b).
    a, b: [T, Hv]; A_log, dt_bias: [Hv]."""
    g = -A_log.float().exp() * F.softplus(a.float() + dt_bias.float())
    beta = torch.sigmoid(b.float())
    return g, beta


def l2norm(x: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    return x / torch.sqrt((x * x).sum(-1, keepdim=True) + eps)


def causal_conv1d_prefill(x: torch.Tensor, weight: torch.Tensor,
                          conv_state: torch.Tensor,
                          has_initial_state: bool) -> torch.Tensor:
    """Depthwise causal conv over one sequence chunk with state carry.

    x: [T, C]; weight: [C, K]; conv_state: [C, K-1] (updated in place).
    Returns silu(conv(x)) [T, C].
    """
    T, C = x.shape
    K = weight.shape[1]
    xt = x.float().T                       # [C, T]
    if has_initial_state:
        ctx = torch.cat([conv_state.float(), xt], dim=1)
    else:
        ctx = torch.cat([torch.zeros(C, K - 1, dtype=torch.float32,
                                     device=x.device), xt], dim=1)
    out = F.conv1d(ctx.unsqueeze(0), weight.float().unsqueeze(1),
                   groups=C).squeeze(0)    # [C, T]
    # save the last K-1 inputs for the next chunk
    conv_state.copy_(ctx[:, -(K - 1):].to(conv_state.dtype))
    return F.silu(out).T.to(x.dtype)


def causal_conv1d_update(x: torch.Tensor, weight: torch.Tensor,
                         conv_state: torch.Tensor) -> torch.Tensor:
    """Single-token decode step. x: [C]; conv_state [C, K-1] rolls."""
    K = weight.shape[1]
    ctx = torch.cat([conv_state.float(), x.float().unsqueeze(1)], dim=1)
    out = (ctx * weight.float()).sum(-1)
    conv_state.copy_(ctx[:, 1:].to(conv_state.dtype))
    return F.silu(out).to(x.dtype)


def gated_delta_rule(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     g: torch.Tensor, beta: torch.Tensor, scale: float,
                     state: torch.Tensor) -> torch.Tensor:
    """Sequential gated delta rule over one sequence.

    q, k: [T, Hk, Dk]; v: [T, Hv, Dv]; g, beta: [T, Hv];
    state: [Hv, Dv, Dk] fp32, updated IN PLACE. Returns o [T, Hv, Dv].
    """
    T, Hk, Dk = q.shape
    Hv, Dv = v.shape[1], v.shape[2]
    G = Hv // Hk
    qn = (l2norm(q.float()) * scale).repeat_interleave(G, dim=1)
    kn = l2norm(k.float()).repeat_interleave(G, dim=1)   # [T, Hv, Dk]
    vf = v.float()
    S = state.float()
    outs = []
    for t in range(T):
        S = S * g[t].exp().view(Hv, 1, 1)
        kt = kn[t]                                        # [Hv, Dk]
        u = vf[t] - torch.einsum("hvk,hk->hv", S, kt)
        u = u * beta[t].unsqueeze(-1)
        S = S + torch.einsum("hv,hk->hvk", u, kt)
        outs.append(torch.einsum("hvk,hk->hv", S, qn[t]))
    state.copy_(S.to(state.dtype))
    return torch.stack(outs).to(v.dtype)


def gated_delta_rule_chunked(q: torch.Tensor, k: torch.Tensor,
                             v: torch.Tensor, g: torch.Tensor,
                             beta: torch.Tensor, scale: float,
                             state: torch.Tensor,
                             chunk: int = 64) -> torch.Tensor:
    """Chunk-parallel gated delta rule — same contract as
    :func:`gated_delta_rule` (state updated IN PLACE), O(T/C) sequential
    steps instead of O(T). The round-2 gfx950 kernel consumes exactly
    this formulation (reference fla/chunk*.py lineage).

    Within a chunk (b = cumsum(g), decays B_t = exp(b_t)):
      u_t = β_t v_t − β_t B_t S_0 k_t − Σ_{i<t} A[t,i] u_i,
        A[t,i] = β_t exp(b_t − b_i) (k_t·k_i)
      => U = (I + A)^{-1} (diag(β)V − diag(β B) K S_0ᵀ)   [unit-lower
         triangular solve — the WY transform]
      O = diag(B) Q S_0ᵀ + M∘(exp(b_t − b_i)(q_t·k_i)) U   (i <= t)
      S_C = B_C S_0 + Σ_i exp(b_C − b_i) u_i k_iᵀ
    All exponent differences are <= 0 (g <= 0), so every term is
    numerically bounded."""
    T, Hk, Dk = q.shape
    Hv, Dv = v.shape[1], v.shape[2]
    G = Hv // Hk
    qn = (l2norm(q.float()) * scale).repeat_interleave(G, dim=1)
    kn = l2norm(k.float()).repeat_interleave(G, dim=1)    # [T, Hv, Dk]
    vf = v.float()
    gf = g.float()
    bf = beta.float()
    S = state.float()                                     # [Hv, Dv, Dk]
    outs = []
    for s in range(0, T, chunk):
        e = min(s + chunk, T)
        C = e - s
        Q = qn[s:e].permute(1, 0, 2)                      # [Hv, C, Dk]
        K = kn[s:e].permute(1, 0, 2)
        V = vf[s:e].permute(1, 0, 2)                      # [Hv, C, Dv]
        b = gf[s:e].cumsum(0).t()                         # [Hv, C]
        bt = bf[s:e].t()                                  # [Hv, C]
        B = b.exp()                                       # [Hv, C]
        # A[t,i] = β_t exp(b_t−b_i) (k_t·k_i), strictly lower
        kk = torch.einsum("htd,hid->hti", K, K)
        dec = (b.unsqueeze(-1) - b.unsqueeze(1)).tril(-1).exp()
        A = (bt.unsqueeze(-1) * dec * kk).tril(-1)
        M = bt.unsqueeze(-1) * (
            V - B.unsqueeze(-1) * torch.einsum("htd,hvd->htv", K, S))
        eye = torch.eye(C, device=A.device).expand_as(A)
        U = torch.linalg.solve_triangular(eye + A, M, upper=False,
                                          unitriangular=True)
        # O = diag(B) Q S0^T + tril(exp(b_t-b_i) q·k) U  (inclusive)
        qk = torch.einsum("htd,hid->hti", Q, K)
        deci = (b.unsqueeze(-1) - b.unsqueeze(1)).tril().exp()
        att = (deci * qk).tril()
        O = B.unsqueeze(-1) * torch.einsum("htd,hvd->htv", Q, S) + \
            torch.einsum("hti,hiv->htv", att, U)
        outs.append(O.permute(1, 0, 2))
        # carry: S = B_C S0 + Σ_i exp(b_C − b_i) u_i k_i^T
        wC = (b[:, C - 1:].expand_as(b) - b).exp()        # [Hv, C]
        S = B[:, C - 1].view(Hv, 1, 1) * S + \
            torch.einsum("htv,htd->hvd", wC.unsqueeze(-1) * U, K)
    state.copy_(S.to(state.dtype))
    return torch.cat(outs).to(v.dtype)


def gated_delta_rule_chunked_batched(
        q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
        g: torch.Tensor, beta: torch.Tensor, scale: float,
        states: torch.Tensor, chunk: int = 256) -> torch.Tensor:
    """Chunk-parallel WY delta rule over a PADDED batch of sequences.

    q, k: [B, T, Hk, Dk]; v: [B, T, Hv, Dv]; g, beta: [B, T, Hv];
    states: [B, Hv, Dv, Dk] fp32 (updated IN PLACE). Padding rows MUST
    carry beta = 0 and g = 0: a zero-beta token has U row 0, so it
    neither updates the state nor contributes to any output — the
    padded math is exactly the ragged math. Returns o [B, T, Hv, Dv].

    This removes the engine's per-sequence python loop (B x ~20
    launches per chunk); with chunk=256 a 1k-token prompt runs 4
    sequential chunk steps instead of 16.
    """
    B, T, Hk, Dk = q.shape
    Hv, Dv = v.shape[2], v.shape[3]
    G = Hv // Hk
    qn = (l2norm(q.float()) * scale).repeat_interleave(G, dim=2)
    kn = l2norm(k.float()).repeat_interleave(G, dim=2)  # [B, T, Hv, Dk]
    vf = v.float()
    gf = g.float()
    bf = beta.float()
    S = states.float()                                  # [B, Hv, Dv, Dk]
    outs = []
    for s in range(0, T, chunk):
        e = min(s + chunk, T)
        C = e - s
        Q = qn[:, s:e].permute(0, 2, 1, 3)              # [B, Hv, C, Dk]
        K = kn[:, s:e].permute(0, 2, 1, 3)
        V = vf[:, s:e].permute(0, 2, 1, 3)              # [B, Hv, C, Dv]
        b = gf[:, s:e].cumsum(1).transpose(1, 2)        # [B, Hv, C]
        bt = bf[:, s:e].transpose(1, 2)                 # [B, Hv, C]
        Bd = b.exp()
        kk = torch.einsum("bhtd,bhid->bhti", K, K)
        dec = (b.unsqueeze(-1) - b.unsqueeze(-2)).tril(-1).exp()
        A = (bt.unsqueeze(-1) * dec * kk).tril(-1)
        M = bt.unsqueeze(-1) * (
            V - Bd.unsqueeze(-1) * torch.einsum("bhtd,bhvd->bhtv", K, S))
        eye = torch.eye(C, device=A.device).expand_as(A)
        U = torch.linalg.solve_triangular(eye + A, M, upper=False,
                                          unitriangular=True)
        qk = torch.einsum("bhtd,bhid->bhti", Q, K)
        deci = (b.unsqueeze(-1) - b.unsqueeze(-2)).tril().exp()
        att = (deci * qk).tril()
        O = Bd.unsqueeze(-1) * torch.einsum("bhtd,bhvd->bhtv", Q, S) + \
            torch.einsum("bhti,bhiv->bhtv", att, U)
        outs.append(O.permute(0, 2, 1, 3))
        wC = (b[..., C - 1:].expand_as(b) - b).exp()    # [B, Hv, C]
        S = Bd[..., C - 1].view(B, Hv, 1, 1) * S + \
            torch.einsum("bhtv,bhtd->bhvd", wC.unsqueeze(-1) * U, K)
    states.copy_(S.to(states.dtype))
    return torch.cat(outs, dim=1).to(v.dtype)


def rmsnorm_gated(x: torch.Tensor, z: torch.Tensor, weight: torch.Tensor,
                  eps: float) -> torch.Tensor:
    """out = rmsnorm(x) * w * silu(z)  (norm applied before the gate;
    reference RMSNormGated with norm_before_gate=True). x, z: [T, D]."""
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    n = xf * torch.rsqrt(var + eps) * weight.float()
    return (n * F.silu(z.float())).to(x.dtype)
