"""Causal (flash-style) attention: CDNA4 HIP kernel + sdpa/eager references.

Replaces the reference's TE DotProductAttention / flash-attn / FlexAttention
backends with one in-tree HIP flash kernel
(csrc/flash_attn.hip): BSHD layout, GQA, online softmax, MFMA 16x16 tiles,
LDS-staged K/V (guide Appendix B "Fused attention prefill").

Shapes: q [B, S, Hq, D], k/v [B, S, Hk, D] with Hq % Hk == 0.
Returns o [B, S, Hq, D].
"""

from __future__ import annotations

import math

import torch

from automodel_amd.ops._backend import hip_ops


def attention_ref(q, k, v, causal: bool = True, scale: float | None = None):
    """Eager fp32 reference used by kernel parity tests."""
    B, S, Hq, D = q.shape
    Hk = k.shape[2]
    scale = scale or 1.0 / math.sqrt(D)
    qf = q.permute(0, 2, 1, 3).float()              # B,Hq,S,D
    kf = k.permute(0, 2, 1, 3).float()
    vf = v.permute(0, 2, 1, 3).float()
    if Hk != Hq:
        rep = Hq // Hk
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        mask = torch.ones(S, S, dtype=torch.bool, device=q.device).tril()
        scores = scores.masked_fill(~mask, float("-inf"))
    p = torch.softmax(scores, dim=-1)
    o = torch.matmul(p, vf)
    return o.permute(0, 2, 1, 3).to(q.dtype)


def _sdpa(q, k, v, causal: bool, scale: float | None):
    qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))   # B,H,S,D
    o = torch.nn.functional.scaled_dot_product_attention(
        qt, kt, vt, is_causal=causal, scale=scale, enable_gqa=q.shape[2] != k.shape[2]
    )
    return o.transpose(1, 2)


def _attention_bwd_composite(do, q, k, v, o, lse, causal: bool, scale: float,
                             q_chunk: int = 1024):
    """Backward as a chain of hipBLASLt GEMMs, q-chunked so the score matrix is
    never fully materialized. Used until the fused HIP backward kernel lands;
    all FLOPs ride MFMA through the GEMM library.

    do/q/o: [B,S,Hq,D]; k/v: [B,S,Hk,D]; lse: [B,Hq,S] (natural log).
    """
    B, S, Hq, D = q.shape
    Hk = k.shape[2]
    rep = Hq // Hk
    # [B*Hq, S, D] views
    qb = q.permute(0, 2, 1, 3).reshape(B * Hq, S, D)
    dob = do.permute(0, 2, 1, 3).reshape(B * Hq, S, D)
    ob = o.permute(0, 2, 1, 3).reshape(B * Hq, S, D)
    kb = k.permute(0, 2, 1, 3).repeat_interleave(rep, dim=1).reshape(B * Hq, S, D)
    vb = v.permute(0, 2, 1, 3).repeat_interleave(rep, dim=1).reshape(B * Hq, S, D)
    lse_b = lse.reshape(B * Hq, S)

    delta = (dob.float() * ob.float()).sum(-1)            # [BH, S]
    dq = torch.empty_like(qb)
    dk_acc = torch.zeros_like(kb, dtype=torch.float32)
    dv_acc = torch.zeros_like(vb, dtype=torch.float32)

    kt = kb.transpose(1, 2)                               # [BH, D, S]
    vt = vb.transpose(1, 2)
    arange_k = torch.arange(S, device=q.device)
    for s0 in range(0, S, q_chunk):
        s1 = min(s0 + q_chunk, S)
        qc = qb[:, s0:s1]                                  # [BH, c, D]
        kv_end = s1 if causal else S
        scores = torch.bmm(qc, kt[:, :, :kv_end]).float() * scale
        p = torch.exp(scores - lse_b[:, s0:s1, None])      # [BH, c, kv]
        if causal:
            mask = arange_k[None, None, :kv_end] > torch.arange(s0, s1, device=q.device)[None, :, None]
            p = p.masked_fill(mask, 0.0)
        pb = p.to(q.dtype)
        dp = torch.bmm(dob[:, s0:s1], vt[:, :, :kv_end]).float()
        ds = (p * (dp - delta[:, s0:s1, None]) * scale).to(q.dtype)
        dq[:, s0:s1] = torch.bmm(ds, kb[:, :kv_end])
        dk_acc[:, :kv_end] += torch.bmm(ds.transpose(1, 2), qc).float()
        dv_acc[:, :kv_end] += torch.bmm(pb.transpose(1, 2), dob[:, s0:s1]).float()

    dqo = dq.reshape(B, Hq, S, D).permute(0, 2, 1, 3).contiguous()
    dk4 = dk_acc.reshape(B, Hk, rep, S, D).sum(2)
    dv4 = dv_acc.reshape(B, Hk, rep, S, D).sum(2)
    dko = dk4.permute(0, 2, 1, 3).to(k.dtype).contiguous()
    dvo = dv4.permute(0, 2, 1, 3).to(v.dtype).contiguous()
    return dqo, dko, dvo


class _FlashAttnHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal: bool, scale: float):
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        o, lse = hip_ops().flash_attn_fwd(q, k, v, scale, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal, ctx.scale = causal, scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = hip_ops().flash_attn_bwd(
            do.contiguous(), q, k, v, o, lse, ctx.scale, ctx.causal
        )
        return dq, dk, dv, None, None


def flash_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    scale: float | None = None,
    backend: str = "hip",
) -> torch.Tensor:
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    if backend == "hip" and q.is_cuda:
        return _FlashAttnHip.apply(q, k, v, causal, scale)
    if backend == "eager":
        return attention_ref(q, k, v, causal, scale)
    return _sdpa(q, k, v, causal, scale)
