"""Unified multimodal (understanding + generation) model — BAGEL-class.

Reference behavior: nemo_automodel/components/models/bagel (unified
understanding CE + visual-generation flow-matching MSE over packed
mixed-modality sequences; recipes/multimodal/finetune.py:688 consumes
``dict(ce=..., mse=...)`` plus token counts). MI355X-native composition from
in-tree parts: the VLM vision tower feeds understanding soft tokens, a small
conv VAE encodes generation targets to latents, generation tokens ride the
SAME llama-style trunk with an adaLN-free timestep embedding added, and a
flow head predicts rectified-flow velocity at the gen positions.
"""

from __future__ import annotations

from dataclasses import dataclass, field

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM
from automodel_amd.models.vlm.model import VisionConfig, VisionTower


def timestep_embedding(t: torch.Tensor, dim: int) -> torch.Tensor:
    half = dim // 2
    freqs = torch.exp(-torch.arange(half, device=t.device).float()
                      * (torch.log(torch.tensor(10000.0)) / half))
    args = t.float()[:, None] * freqs[None]
    return torch.cat([args.cos(), args.sin()], dim=-1)


class TinyVAE(nn.Module):
    """In-tree latent encoder (the reference loads an external VAE sidecar;
    offline we train against this conv encoder's latents, 8x downsample)."""

    def __init__(self, channels: int = 3, latent_dim: int = 16):
        super().__init__()
        self.latent_dim = latent_dim
        self.enc = nn.Sequential(
            nn.Conv2d(channels, 32, 4, 2, 1), nn.SiLU(),
            nn.Conv2d(32, 64, 4, 2, 1), nn.SiLU(),
            nn.Conv2d(64, latent_dim, 4, 2, 1),
        )

    @torch.no_grad()
    def encode(self, images: torch.Tensor) -> torch.Tensor:
        return self.enc(images)


@dataclass
class OmniConfig:
    text: LlamaConfig = field(default_factory=LlamaConfig)
    vision: VisionConfig = field(default_factory=VisionConfig)
    latent_dim: int = 16
    gen_patch: int = 2           # latent patch size for gen tokens
    image_token_id: int = 3
    gen_token_id: int = 4

    def __post_init__(self):
        if isinstance(self.text, dict):
            self.text = LlamaConfig(**self.text)
        if isinstance(self.vision, dict):
            self.vision = VisionConfig(**self.vision)


class OmniForUnifiedMultimodal(nn.Module):
    config_class = OmniConfig

    def __init__(self, config: OmniConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = OmniConfig(**config)
        self.config = config
        H = config.text.hidden_size
        self.visual = VisionTower(config.vision)
        self.projector = nn.Sequential(
            nn.Linear(config.vision.hidden_size, H), nn.GELU(), nn.Linear(H, H))
        self.language_model = LlamaForCausalLM(config.text, backend=backend)
        gen_in = config.latent_dim * config.gen_patch ** 2
        self.gen_in_proj = nn.Linear(gen_in, H)
        self.time_embed = nn.Sequential(nn.Linear(H, H), nn.SiLU(), nn.Linear(H, H))
        self.flow_head = nn.Sequential(nn.Linear(H, H), nn.SiLU(),
                                       nn.Linear(H, gen_in))
        self.vae = TinyVAE(latent_dim=config.latent_dim)
        self.loss_fn = None

    def freeze_vision_tower(self) -> None:
        for p in self.visual.parameters():
            p.requires_grad_(False)

    def _patchify(self, z: torch.Tensor) -> torch.Tensor:
        B, C, Hh, Ww = z.shape
        p = self.config.gen_patch
        z = z.view(B, C, Hh // p, p, Ww // p, p)
        return z.permute(0, 2, 4, 1, 3, 5).reshape(B, (Hh // p) * (Ww // p), -1)

    def forward(self, input_ids, pixel_values=None, gen_images=None,
                timesteps=None, labels=None, **_):
        """Packed forward. ``input_ids`` carries image_token_id slots for
        understanding patches and gen_token_id slots for generation latents.
        Returns dict(ce=sum CE, ce_tokens, mse=sum per-token MSE, mse_tokens).
        """
        lm = self.language_model
        embeds = lm.model.embed_tokens(input_ids)
        B, S = input_ids.shape
        if pixel_values is not None and pixel_values.numel() > 0:
            img = self.projector(self.visual(pixel_values.to(embeds.dtype)))
            img = img.reshape(-1, img.shape[-1])
            mask = input_ids == self.config.image_token_id
            assert int(mask.sum()) == img.shape[0], "image slots != patches"
            embeds = embeds.clone()
            embeds[mask] = img.to(embeds.dtype)

        gen_mask = input_ids == self.config.gen_token_id
        target_v = None
        if gen_images is not None and gen_images.numel() > 0:
            with torch.no_grad():
                z1 = self._patchify(self.vae.encode(gen_images.float()))
            z1 = z1.to(embeds.dtype)
            nB = z1.shape[0]
            if timesteps is None:
                timesteps = torch.rand(nB, device=embeds.device)
            noise = torch.randn_like(z1.float())
            t = timesteps.view(nB, 1, 1).float()
            zt = ((1.0 - t) * noise + t * z1.float())
            target_v = (z1.float() - noise).reshape(-1, z1.shape[-1])
            gen_tok = self.gen_in_proj(zt.to(embeds.dtype))
            temb = self.time_embed(
                timestep_embedding(timesteps, embeds.shape[-1]).to(embeds.dtype))
            gen_tok = gen_tok + temb[:, None]
            if not gen_mask.any():
                target_v = None
            else:
                assert int(gen_mask.sum()) == gen_tok.shape[0] * gen_tok.shape[1], \
                    "gen slots != latent patches"
                embeds = embeds if pixel_values is not None else embeds.clone()
                embeds = embeds.clone()
                embeds[gen_mask] = gen_tok.reshape(-1, gen_tok.shape[-1])

        x = embeds
        cos, sin = lm.model.rope_cos[:S].float(), lm.model.rope_sin[:S].float()
        for layer in lm.model.layers:
            x = layer(x, cos, sin)
        x = lm.model.norm(x)

        out = {}
        if labels is not None:
            ce_mask = labels != -100
            out["ce_tokens"] = int(ce_mask.sum())
            if out["ce_tokens"]:
                logits = lm.lm_head(x[ce_mask])
                out["ce"] = F.cross_entropy(logits.float(), labels[ce_mask],
                                            reduction="sum")
            else:
                out["ce"] = x.sum() * 0.0
        if target_v is not None:
            v_pred = self.flow_head(x[gen_mask])
            out["mse"] = ((v_pred.float() - target_v) ** 2).mean(dim=-1).sum()
            out["mse_tokens"] = int(gen_mask.sum())
        return out

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
        self.language_model.init_weights(device=device)
        for m in (*self.visual.modules(), *self.projector.modules(),
                  *self.gen_in_proj.modules(), *self.time_embed.modules(),
                  *self.flow_head.modules(), *self.vae.modules()):
            if isinstance(m, (nn.Linear, nn.Conv2d)):
                nn.init.normal_(m.weight, std=0.02)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.LayerNorm):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
            elif isinstance(m, nn.MultiheadAttention):
                # raw in_proj parameters are NOT nn.Linear — without this
                # they survive to_empty() as garbage (found via omni recipe
                # NaNs; the out_proj IS a Linear and is covered above)
                nn.init.normal_(m.in_proj_weight, std=0.02)
                if m.in_proj_bias is not None:
                    nn.init.zeros_(m.in_proj_bias)
        if not self.visual.pos_embed.is_meta:
            nn.init.normal_(self.visual.pos_embed, std=0.02)

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
