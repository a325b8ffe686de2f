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


# (Dqk, Dv) pairs the HIP kernels are instantiated for (flash_attn.hip);
# (192, 128) is the MLA shape (qk = nope+rope, v = v_head_dim).
_FLASH_DIMS = (64, 96, 128, 192, 256)
_FLASH_PAIRS = {(64, 64), (96, 96), (128, 128), (192, 128), (192, 192), (256, 256)}


def _target_dims(dqk: int, dv: int) -> tuple[int, int]:
    """Smallest supported (Dqk, Dv) kernel pair >= the given dims (other
    dims are zero-padded by the wrapper: padding qk dims leaves scores
    unchanged, padded V columns are sliced off the output)."""
    if (dqk, dv) in _FLASH_PAIRS:
        return dqk, dv
    tq = next((d for d in _FLASH_DIMS if d >= dqk), None)
    tv = next((d for d in _FLASH_DIMS if d >= dv), None)
    if tq is None or tv is None:
        raise ValueError(f"head dims ({dqk},{dv}) exceed flash kernel max 256")
    if (tq, tv) in _FLASH_PAIRS:
        return tq, tv
    t = max(tq, tv)
    return t, t


def _pad_head(t: torch.Tensor, target: int) -> torch.Tensor:
    d = t.shape[-1]
    return t if d == target else torch.nn.functional.pad(t, (0, target - d))


class _FlashAttnHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal: bool, scale: float, q_start: int,
                doc_start=None, doc_end=None):
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        o, lse = hip_ops().flash_attn_fwd(q, k, v, scale, causal, q_start, doc_start)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal, ctx.scale, ctx.q_start = causal, scale, q_start
        ctx.docs = (doc_start, doc_end)
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        ds, de = ctx.docs
        dq, dk, dv = hip_ops().flash_attn_bwd(
            do.contiguous(), q, k, v, o, lse, ctx.scale, ctx.causal, ctx.q_start,
            ds, de,
        )
        return dq, dk, dv, None, None, None, None, None


def _flash_hip(q, k, v, causal, scale, q_start, doc_start=None, doc_end=None):
    """Dispatch to the HIP kernel, zero-padding head dims to a supported
    kernel pair and (for plain causal) sequence lengths to the kernel's
    tile granularity — pad/slice sit outside the Function so autograd
    routes gradients through them."""
    dqk, dv = q.shape[-1], v.shape[-1]
    tq, tv = _target_dims(dqk, dv)
    if tq != dqk or tv != dv:
        q, k, v = _pad_head(q, tq), _pad_head(k, tq), _pad_head(v, tv)
    sq, skv = q.shape[1], k.shape[1]
    pad_q = (-sq) % 128
    pad_kv = (-skv) % 128 if doc_start is None else (-skv) % 32
    if pad_q or pad_kv:
        # safe only for plain causal from position 0: padded KV rows sit at
        # positions above every real q row, so the causal mask excludes them
        assert causal and q_start == 0 and doc_start is None, \
            "flash: Sq%128/Skv%32 required for q_start/varlen/non-causal calls"
        F = torch.nn.functional.pad
        if pad_q:
            q = F(q, (0, 0, 0, 0, 0, pad_q))
        if pad_kv:
            k = F(k, (0, 0, 0, 0, 0, pad_kv))
            v = F(v, (0, 0, 0, 0, 0, pad_kv))
    o = _FlashAttnHip.apply(q, k, v, causal, scale, q_start, doc_start, doc_end)
    if pad_q:
        o = o[:, :sq]
    return o[..., :dv] if tv != dv else o


_VARLEN_CU: torch.Tensor | None = None


def set_varlen_context(cu_seqlens: torch.Tensor | None) -> None:
    """Install cu_seqlens for packed (THD) batches — attention then applies
    block-diagonal causal masking per document (reference THD runtime,
    distributed/thd_utils.py:85)."""
    global _VARLEN_CU
    _VARLEN_CU = cu_seqlens


def doc_bounds_from_cu(cu_seqlens: torch.Tensor, total: int) -> tuple[torch.Tensor, torch.Tensor]:
    """Per-token document bounds for the varlen kernels: doc_start[t] /
    doc_end[t] of the packed document containing token t. Tokens in
    [cu[-1], total) (seq-length padding) form their own trailing doc."""
    cu = cu_seqlens.to(torch.long)
    lengths = cu[1:] - cu[:-1]
    ds = torch.repeat_interleave(cu[:-1], lengths)
    de = torch.repeat_interleave(cu[1:], lengths)
    t_real = int(cu[-1])
    if t_real < total:
        pad = total - t_real
        ds = torch.cat([ds, torch.full((pad,), t_real, dtype=torch.long, device=ds.device)])
        de = torch.cat([de, torch.full((pad,), total, dtype=torch.long, device=de.device)])
    return ds.to(torch.int32).contiguous(), de.to(torch.int32).contiguous()


def flash_attention_varlen(q, k, v, cu_seqlens: torch.Tensor, scale: float | None = None,
                           backend: str = "hip") -> torch.Tensor:
    """Packed-sequence attention: q/k/v [1, T, H, D], cu_seqlens int32 [n+1].

    GPU path: ONE kernel launch per batch — per-token doc bounds gate the
    causal mask inside the HIP kernel and clip KV-tile ranges per block
    (replaces the round-1 per-document launch loop with its host-side
    cu_seqlens.tolist() sync; reference THD runtime thd_utils.py:85).
    CPU path: one sdpa call with a dense block-causal mask.
    """
    assert q.shape[0] == 1, "varlen expects a packed THD batch [1, T, H, D]"
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    if backend == "hip" and q.is_cuda:
        T = q.shape[1]
        pad = (-T) % 128
        Tp = T + pad
        cu_dev = cu_seqlens.to(q.device)
        ds, de = doc_bounds_from_cu(cu_dev, Tp)
        if pad:
            q = torch.nn.functional.pad(q, (0, 0, 0, 0, 0, pad))
            k = torch.nn.functional.pad(k, (0, 0, 0, 0, 0, pad))
            v = torch.nn.functional.pad(v, (0, 0, 0, 0, 0, pad))
        o = _flash_hip(q, k, v, True, scale, 0, ds, de)
        return o[:, :T]
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
        return _flash_hip(q, k, v, causal, scale, q_start)
    if backend == "eager":
        return attention_ref(q, k, v, causal, scale, q_start)
    return _sdpa(q, k, v, causal, scale, q_start)
