"""Rotary position embedding: fused HIP q+k application, host-precomputed tables.

Replaces the reference's fused-RoPE backends (TE/QuACK/Liger).
Convention matches HF Llama rotate-half: y = x*cos + rotate_half(x)*sin with
rotate_half([x1, x2]) = [-x2, x1] over the two halves of head_dim. cos/sin are
precomputed on host (guide Appendix B: never per-element device trig) with
shape [S, head_dim] (half-table duplicated, fp32).
"""

from __future__ import annotations

import torch

from automodel_amd.ops._backend import hip_ops


def build_rope_cache(
    head_dim: int,
    max_seq_len: int,
    base: float = 10000.0,
    scaling: dict | None = None,
    device=None,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Return (cos, sin) of shape [max_seq_len, head_dim], fp32.

    ``scaling`` supports HF rope_scaling dicts: llama3-style
    {rope_type: "llama3", factor, low_freq_factor, high_freq_factor,
    original_max_position_embeddings} and linear {rope_type: "linear", factor}.
    """
    inv_freq = 1.0 / (base ** (torch.arange(0, head_dim, 2, dtype=torch.float32) / head_dim))
    if scaling:
        rope_type = scaling.get("rope_type", scaling.get("type", "default"))
        if rope_type == "linear":
            inv_freq = inv_freq / float(scaling["factor"])
        elif rope_type == "yarn":
            # YaRN (public formulas; HF _compute_yarn_parameters): NTK-by-parts
            # interpolation with a wavelength ramp + attention temperature.
            import math as _m

            factor = float(scaling.get("factor", 1.0))
            beta_fast = float(scaling.get("beta_fast", 32.0))
            beta_slow = float(scaling.get("beta_slow", 1.0))
            orig = float(scaling.get("original_max_position_embeddings", 4096))
            attn_factor = scaling.get("attention_factor")
            mscale_all = scaling.get("mscale")  # deepseek-style override
            dim = head_dim

            def find_dim(num_rot):
                return (dim * _m.log(orig / (num_rot * 2 * _m.pi))) / (2 * _m.log(base))

            low, high = find_dim(beta_fast), find_dim(beta_slow)
            if scaling.get("truncate", True):
                low, high = _m.floor(low), _m.ceil(high)
            low, high = max(low, 0), min(high, dim - 1)
            ramp = ((torch.arange(dim // 2, dtype=torch.float32) - low)
                    / max(high - low, 1e-3)).clamp(0.0, 1.0)
            extrap_mask = 1.0 - ramp
            inv_freq = inv_freq * (1 - extrap_mask) / factor + inv_freq * extrap_mask
            if attn_factor is not None:
                attention_scaling = float(attn_factor)
            elif mscale_all is not None:
                attention_scaling = 0.1 * float(mscale_all) * _m.log(factor) + 1.0
            else:
                attention_scaling = 0.1 * _m.log(factor) + 1.0
        elif rope_type == "llama3":
            factor = float(scaling["factor"])
            low = float(scaling.get("low_freq_factor", 1.0))
            high = float(scaling.get("high_freq_factor", 4.0))
            orig = float(scaling.get("original_max_position_embeddings", 8192))
            wavelen = 2 * torch.pi / inv_freq
            smooth = ((orig / wavelen - low) / (high - low)).clamp(0.0, 1.0)
            scaled = inv_freq / factor
            # three bands (HF llama3 rope): short wavelengths unchanged,
            # long wavelengths fully scaled, middle band interpolated
            inv_freq = torch.where(
                wavelen < orig / high,                       # high-freq: keep
                inv_freq,
                torch.where(
                    wavelen > orig / low,                    # low-freq: /factor
                    scaled,
                    (1 - smooth) * scaled + smooth * inv_freq,
                ),
            )
    t = torch.arange(max_seq_len, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)
    emb = torch.cat([freqs, freqs], dim=-1)
    scale = locals().get("attention_scaling", 1.0)
    cos, sin = emb.cos() * scale, emb.sin() * scale
    if device is not None:
        cos, sin = cos.to(device), sin.to(device)
    return cos, sin


def _rotate_half(x: torch.Tensor) -> torch.Tensor:
    x1, x2 = x.chunk(2, dim=-1)
    return torch.cat([-x2, x1], dim=-1)


def apply_rope_ref(
    q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor
) -> tuple[torch.Tensor, torch.Tensor]:
    """q,k: [B, S, H, D]; cos/sin: [S, D] (or [B, S, D] for m-rope) fp32."""
    if cos.dim() == 3:                     # batch-shaped tables (qwen2-vl m-rope)
        c = cos.to(q.dtype)[:, :, None, :]
        s = sin.to(q.dtype)[:, :, None, :]
    else:
        c = cos.to(q.dtype)[None, :, None, :]
        s = sin.to(q.dtype)[None, :, None, :]
    return q * c + _rotate_half(q) * s, k * c + _rotate_half(k) * s


class _RopeHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, cos, sin):
        q, k = q.contiguous(), k.contiguous()
        qo, ko = hip_ops().rope_fwd(q, k, cos, sin, False)
        ctx.save_for_backward(cos, sin)
        return qo, ko

    @staticmethod
    def backward(ctx, dq, dk):
        cos, sin = ctx.saved_tensors
        # d/dx of rotate is rotate by -theta: cos stays, sin negates.
        dqi, dki = hip_ops().rope_fwd(dq.contiguous(), dk.contiguous(), cos, sin, True)
        return dqi, dki, None, None


def apply_rope(
    q: torch.Tensor,
    k: torch.Tensor,
    cos: torch.Tensor,
    sin: torch.Tensor,
    backend: str = "hip",
) -> tuple[torch.Tensor, torch.Tensor]:
    """Apply RoPE to q [B,S,Hq,D] and k [B,S,Hk,D] with position-sliced tables."""
    if backend == "hip" and q.is_cuda and cos.dim() == 2:
        return _RopeHip.apply(q, k, cos, sin)
    return apply_rope_ref(q, k, cos, sin)
