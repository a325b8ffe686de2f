"""DiT (diffusion transformer) for image flow-matching.

Reference behavior: nemo_automodel/recipes/diffusion/train.py + _diffusers/
flow-matching adapters (FLUX/Wan-style training loops). This is the
MI355X-native in-tree model for the diffusion recipe: patchify -> N
adaLN-modulated transformer blocks (torch SDPA attention — non-causal,
small head dims) -> unpatchify, trained with rectified flow
(recipes/diffusion/train.py).
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn


@dataclass
class DiTConfig:
    image_size: int = 32
    patch_size: int = 4
    in_channels: int = 3
    hidden_size: int = 256
    num_hidden_layers: int = 4
    num_attention_heads: int = 4
    mlp_ratio: float = 4.0

    @property
    def n_patches(self) -> int:
        return (self.image_size // self.patch_size) ** 2


def timestep_embedding(t: torch.Tensor, dim: int) -> torch.Tensor:
    half = dim // 2
    freqs = torch.exp(-math.log(10000.0) * torch.arange(half, device=t.device) / half)
    ang = t[:, None].float() * freqs[None]
    return torch.cat([ang.cos(), ang.sin()], dim=-1)


class DiTBlock(nn.Module):
    """Pre-LN transformer block with adaLN-zero conditioning on t."""

    def __init__(self, cfg: DiTConfig):
        super().__init__()
        h = cfg.hidden_size
        self.norm1 = nn.LayerNorm(h, elementwise_affine=False)
        self.attn = nn.MultiheadAttention(h, cfg.num_attention_heads, batch_first=True)
        self.norm2 = nn.LayerNorm(h, elementwise_affine=False)
        inner = int(h * cfg.mlp_ratio)
        self.mlp = nn.Sequential(nn.Linear(h, inner), nn.GELU(), nn.Linear(inner, h))
        self.ada = nn.Linear(h, 6 * h)
        nn.init.zeros_(self.ada.weight)
        nn.init.zeros_(self.ada.bias)

    def forward(self, x: torch.Tensor, c: torch.Tensor) -> torch.Tensor:
        s1, b1, g1, s2, b2, g2 = self.ada(c)[:, None].chunk(6, dim=-1)
        h = self.norm1(x) * (1 + s1) + b1
        h, _ = self.attn(h, h, h, need_weights=False)
        x = x + g1 * h
        h = self.mlp(self.norm2(x) * (1 + s2) + b2)
        return x + g2 * h


class DiTForFlowMatching(nn.Module):
    config_class = DiTConfig

    def __init__(self, config: DiTConfig | dict, backend=None):
        super().__init__()
        cfg = config if isinstance(config, DiTConfig) else DiTConfig(**dict(config))
        self.config = cfg
        p, c, h = cfg.patch_size, cfg.in_channels, cfg.hidden_size
        self.patch_embed = nn.Linear(p * p * c, h)
        self.pos_embed = nn.Parameter(torch.zeros(1, cfg.n_patches, h))
        self.t_mlp = nn.Sequential(nn.Linear(h, h), nn.SiLU(), nn.Linear(h, h))
        self.blocks = nn.ModuleList(DiTBlock(cfg) for _ in range(cfg.num_hidden_layers))
        self.final_norm = nn.LayerNorm(h, elementwise_affine=False)
        self.final_ada = nn.Linear(h, 2 * h)
        self.head = nn.Linear(h, p * p * c)
        nn.init.zeros_(self.final_ada.weight)
        nn.init.zeros_(self.final_ada.bias)
        nn.init.zeros_(self.head.weight)
        nn.init.zeros_(self.head.bias)
        nn.init.normal_(self.pos_embed, std=0.02)

    def _patchify(self, x: torch.Tensor) -> torch.Tensor:
        B, C, H, W = x.shape
        p = self.config.patch_size
        x = x.reshape(B, C, H // p, p, W // p, p)
        return x.permute(0, 2, 4, 3, 5, 1).reshape(B, -1, p * p * C)

    def _unpatchify(self, x: torch.Tensor) -> torch.Tensor:
        cfg = self.config
        p, C = cfg.patch_size, cfg.in_channels
        B, N, _ = x.shape
        g = cfg.image_size // p
        x = x.reshape(B, g, g, p, p, C)
        return x.permute(0, 5, 1, 3, 2, 4).reshape(B, C, g * p, g * p)

    def forward(self, xt: torch.Tensor, t: torch.Tensor) -> torch.Tensor:
        """xt [B, C, H, W] noisy image, t [B] in [0, 1] -> velocity field."""
        cfg = self.config
        c = self.t_mlp(timestep_embedding(t, cfg.hidden_size).to(xt.dtype))
        h = self.patch_embed(self._patchify(xt)) + self.pos_embed
        for blk in self.blocks:
            h = blk(h, c)
        s, b = self.final_ada(c)[:, None].chunk(2, dim=-1)
        return self._unpatchify(self.head(self.final_norm(h) * (1 + s) + b))

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to(device)

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
