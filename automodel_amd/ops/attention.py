"""Causal (flash-style) attention: CDNA4 HIP kernel + sdpa/eager references.

Replaces the reference's TE DotProductAttention / flash-attn / FlexAttention
backends with one in-tree HIP flash kernel
(csrc/flash_attn.hip): BSHD layout, GQA, online softmax, MFMA 32x32 tiles,
LDS-staged K/V, fused backward (dKV kv-parallel + dQ q-parallel kernels).

Shapes: q [B, Sq, Hq, D], k/v [B, Skv, Hk, D] with Hq % Hk == 0.
``q_start`` offsets q positions against the kv sequence (context parallelism:
each CP rank attends its q chunks over the gathered KV).
"""

from __future__ import annotations

import math

import torch

from automodel_amd.ops._backend import hip_ops


def _causal_mask(Sq: int, Skv: int, q_start: int, device) -> torch.Tensor:
    qpos = torch.arange(q_start, q_start + Sq, device=device)
    kpos = torch.arange(Skv, device=device)
    return kpos[None, :] <= qpos[:, None]       # [Sq, Skv] True = keep


def attention_ref(q, k, v, causal: bool = True, scale: float | None = None,
                  q_start: int = 0):
    """Eager fp32 reference used by kernel parity tests."""
    B, Sq, Hq, D = q.shape
    Skv, Hk = k.shape[1], k.shape[2]
    scale = scale or 1.0 / math.sqrt(D)
    qf = q.permute(0, 2, 1, 3).float()              # B,Hq,Sq,D
    kf = k.permute(0, 2, 1, 3).float()
    vf = v.permute(0, 2, 1, 3).float()
    if Hk != Hq:
        rep = Hq // Hk
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        scores = scores.masked_fill(~_causal_mask(Sq, Skv, q_start, q.device), float("-inf"))
    p = torch.softmax(scores, dim=-1)
    o = torch.matmul(p, vf)
    return o.permute(0, 2, 1, 3).to(q.dtype)


def sdpa_masked(q, k, v, attn_mask, scale: float | None = None):
    """sdpa with an explicit additive mask — used by the static-cache
    (hipGraph) decode path where causal structure lives in the mask."""
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))
    o = torch.nn.functional.scaled_dot_product_attention(
        qt, kt, vt, attn_mask=attn_mask, scale=scale,
        enable_gqa=q.shape[2] != k.shape[2])
    return o.transpose(1, 2)


def _sdpa(q, k, v, causal: bool, scale: float | None, q_start: int = 0):
    qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))   # B,H,S,D
    kwargs = dict(scale=scale, enable_gqa=q.shape[2] != k.shape[2])
    if causal and (q_start != 0 or q.shape[1] != k.shape[1]):
        mask = _causal_mask(q.shape[1], k.shape[1], q_start, q.device)
        o = torch.nn.functional.scaled_dot_product_attention(qt, kt, vt,
                                                             attn_mask=mask, **kwargs)
    else:
        o = torch.nn.functional.scaled_dot_product_attention(qt, kt, vt,
                                                             is_causal=causal, **kwargs)
    return o.transpose(1, 2)


class _FlashAttnHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal: bool, scale: float, q_start: int):
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        o, lse = hip_ops().flash_attn_fwd(q, k, v, scale, causal, q_start)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal, ctx.scale, ctx.q_start = causal, scale, q_start
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = hip_ops().flash_attn_bwd(
            do.contiguous(), q, k, v, o, lse, ctx.scale, ctx.causal, ctx.q_start
        )
        return dq, dk, dv, None, None, None


_VARLEN_CU: torch.Tensor | None = None


def set_varlen_context(cu_seqlens: torch.Tensor | None) -> None:
    """Install cu_seqlens for packed (THD) batches — attention then applies
    block-diagonal causal masking per document (reference THD runtime,
    distributed/thd_utils.py:85)."""
    global _VARLEN_CU
    _VARLEN_CU = cu_seqlens


def flash_attention_varlen(q, k, v, cu_seqlens: torch.Tensor, scale: float | None = None,
                           backend: str = "hip") -> torch.Tensor:
    """Packed-sequence attention: q/k/v [1, T, H, D], cu_seqlens int32 [n+1].

    GPU path: per-document HIP flash calls with zero-padding to the kernel's
    128-row granularity — appended pad KV sits at positions > every real q
    position, so causal masking excludes it; pad q rows are sliced off.
    CPU path: one sdpa call with a dense block-causal mask.
    """
    assert q.shape[0] == 1, "varlen expects a packed THD batch [1, T, H, D]"
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    bounds = cu_seqlens.tolist()
    if backend == "hip" and q.is_cuda:
        outs = []
        for a, b in zip(bounds[:-1], bounds[1:]):
            L = b - a
            if L == 0:
                continue
            pad = (-L) % 128
            qc, kc, vc = (t[:, a:b] for t in (q, k, v))
            if pad:
                qc = torch.nn.functional.pad(qc, (0, 0, 0, 0, 0, pad))
                kc = torch.nn.functional.pad(kc, (0, 0, 0, 0, 0, pad))
                vc = torch.nn.functional.pad(vc, (0, 0, 0, 0, 0, pad))
            o = _FlashAttnHip.apply(qc, kc, vc, True, scale, 0)
            outs.append(o[:, :L])
        return torch.cat(outs, dim=1)
    from automodel_amd.datasets.llm.packed_sequence import block_causal_mask

    mask = block_causal_mask(cu_seqlens.cpu(), q.shape[1]).to(q.device)
    qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))
    o = torch.nn.functional.scaled_dot_product_attention(
        qt, kt, vt, attn_mask=mask, scale=scale,
        enable_gqa=q.shape[2] != k.shape[2])
    return o.transpose(1, 2)


def flash_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    scale: float | None = None,
    backend: str = "hip",
    q_start: int = 0,
) -> torch.Tensor:
    if _VARLEN_CU is not None and causal and q_start == 0:
        return flash_attention_varlen(q, k, v, _VARLEN_CU, scale, backend)
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    if backend == "hip" and q.is_cuda:
        return _FlashAttnHip.apply(q, k, v, causal, scale, q_start)
    if backend == "eager":
        return attention_ref(q, k, v, causal, scale, q_start)
    return _sdpa(q, k, v, causal, scale, q_start)
