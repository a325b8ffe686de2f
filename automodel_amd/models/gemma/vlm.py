"""Gemma-3 multimodal: SigLIP vision tower + soft-token projector + the
gemma-3 text decoder with image-block bidirectional attention.

Reference behavior: nemo_automodel/components/models/gemma4_unified (wraps
HF gemma multimodal). Implemented directly against public Gemma-3:

  * SigLIP ViT: conv patch embed WITH bias, learned positions, NO class
    token, pre-LN encoder blocks (tanh-GELU MLP), post layernorm;
  * projector: avg-pool the patch grid down to mm_tokens_per_image soft
    tokens, Gemma-RMSNorm, then matmul with mm_input_projection_weight;
  * text: models/gemma Gemma3 stack; image tokens attend BIDIRECTIONALLY
    within their own image block (block_ids overlay), causally otherwise.

HF keys match Gemma3ForConditionalGeneration (parity-tested).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.gemma.model import (
    Gemma3ForCausalLM,
    GemmaConfig,
    GemmaRMSNorm,
)


@dataclass
class SiglipVisionConfig:
    hidden_size: int = 1152
    intermediate_size: int = 4304
    num_hidden_layers: int = 27
    num_attention_heads: int = 16
    image_size: int = 896
    patch_size: int = 14
    num_channels: int = 3
    layer_norm_eps: float = 1e-6

    @property
    def num_patches(self) -> int:
        return (self.image_size // self.patch_size) ** 2


@dataclass
class Gemma3VLMConfig:
    text: GemmaConfig = field(default_factory=GemmaConfig)
    vision: SiglipVisionConfig = field(default_factory=SiglipVisionConfig)
    mm_tokens_per_image: int = 256
    image_token_id: int = 262144

    def __post_init__(self):
        if isinstance(self.text, dict):
            self.text = GemmaConfig(**self.text)
        if isinstance(self.vision, dict):
            self.vision = SiglipVisionConfig(**self.vision)

    @property
    def vocab_size(self):
        return self.text.vocab_size

    @property
    def num_hidden_layers(self):
        return self.text.num_hidden_layers

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Gemma3VLMConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        if "text" in hf and "vision" in hf:   # our own exported config.json
            import dataclasses as _dc

            keys = {f.name for f in _dc.fields(cls)}
            return cls(**{k: v for k, v in hf.items() if k in keys})
        tc = hf.get("text_config", hf.get("text", {}))
        vc = hf.get("vision_config", hf.get("vision", {}))
        return cls(
            text=GemmaConfig.from_hf_gemma3(tc),
            vision=SiglipVisionConfig(
                hidden_size=vc.get("hidden_size", 1152),
                intermediate_size=vc.get("intermediate_size", 4304),
                num_hidden_layers=vc.get("num_hidden_layers", 27),
                num_attention_heads=vc.get("num_attention_heads", 16),
                image_size=vc.get("image_size", 896),
                patch_size=vc.get("patch_size", 14),
                num_channels=vc.get("num_channels", 3),
                layer_norm_eps=vc.get("layer_norm_eps", 1e-6),
            ),
            mm_tokens_per_image=hf.get("mm_tokens_per_image", 256),
            image_token_id=hf.get("image_token_id",
                                  hf.get("image_token_index", 262144)),
        )


class SiglipEncoderLayer(nn.Module):
    def __init__(self, cfg: SiglipVisionConfig):
        super().__init__()
        H = cfg.num_attention_heads
        self.layer_norm1 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.layer_norm2 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        attn = nn.Module()
        attn.q_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        attn.k_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        attn.v_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        attn.out_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.self_attn = attn
        self.num_heads = H
        mlp = nn.Module()
        mlp.fc1 = nn.Linear(cfg.hidden_size, cfg.intermediate_size)
        mlp.fc2 = nn.Linear(cfg.intermediate_size, cfg.hidden_size)
        self.mlp = mlp

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        h = self.layer_norm1(x)
        B, N, _ = h.shape
        a = self.self_attn
        q = a.q_proj(h).view(B, N, self.num_heads, -1).transpose(1, 2)
        k = a.k_proj(h).view(B, N, self.num_heads, -1).transpose(1, 2)
        v = a.v_proj(h).view(B, N, self.num_heads, -1).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v)
        x = x + a.out_proj(o.transpose(1, 2).reshape(B, N, -1))
        h = self.mlp.fc1(self.layer_norm2(x))
        return x + self.mlp.fc2(F.gelu(h, approximate="tanh"))


class SiglipVisionTower(nn.Module):
    def __init__(self, cfg: SiglipVisionConfig):
        super().__init__()
        self.cfg = cfg
        emb = nn.Module()
        emb.patch_embedding = nn.Conv2d(cfg.num_channels, cfg.hidden_size,
                                        kernel_size=cfg.patch_size,
                                        stride=cfg.patch_size, bias=True)
        emb.position_embedding = nn.Embedding(cfg.num_patches, cfg.hidden_size)
        self.embeddings = emb
        enc = nn.Module()
        enc.layers = nn.ModuleList(SiglipEncoderLayer(cfg)
                                   for _ in range(cfg.num_hidden_layers))
        self.encoder = enc
        self.post_layernorm = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        # dormant SigLIP attention-pooling head: never used by gemma-3's
        # forward, kept so checkpoints round-trip with zero dropped keys
        head = nn.Module()
        head.probe = nn.Parameter(torch.zeros(1, 1, cfg.hidden_size))
        head.attention = nn.MultiheadAttention(cfg.hidden_size,
                                               cfg.num_attention_heads,
                                               batch_first=True)
        head.layernorm = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        hm = nn.Module()
        hm.fc1 = nn.Linear(cfg.hidden_size, cfg.intermediate_size)
        hm.fc2 = nn.Linear(cfg.intermediate_size, cfg.hidden_size)
        head.mlp = hm
        self.head = head

    def forward(self, pixel_values: torch.Tensor) -> torch.Tensor:
        x = self.embeddings.patch_embedding(
            pixel_values.to(self.embeddings.patch_embedding.weight.dtype))
        x = x.flatten(2).transpose(1, 2)
        x = x + self.embeddings.position_embedding.weight[None]
        for layer in self.encoder.layers:
            x = layer(x)
        return self.post_layernorm(x)


class Gemma3MultiModalProjector(nn.Module):
    def __init__(self, cfg: Gemma3VLMConfig):
        super().__init__()
        self.mm_input_projection_weight = nn.Parameter(
            torch.zeros(cfg.vision.hidden_size, cfg.text.hidden_size))
        self.mm_soft_emb_norm = GemmaRMSNorm(cfg.vision.hidden_size,
                                             cfg.vision.layer_norm_eps, "torch")
        per_side = cfg.vision.image_size // cfg.vision.patch_size
        tokens_side = int(cfg.mm_tokens_per_image ** 0.5)
        self.per_side = per_side
        self.pool = nn.AvgPool2d(kernel_size=per_side // tokens_side,
                                 stride=per_side // tokens_side)

    def forward(self, feats: torch.Tensor) -> torch.Tensor:
        B, P, H = feats.shape
        grid = feats.transpose(1, 2).reshape(B, H, self.per_side, self.per_side)
        pooled = self.pool(grid).flatten(2).transpose(1, 2)
        return self.mm_soft_emb_norm(pooled) @ self.mm_input_projection_weight


def image_block_ids(input_ids: torch.Tensor, image_token_id: int) -> torch.Tensor:
    """HF get_block_sequence_ids_for_mask: per-token image-group id, -1 for
    text; consecutive image tokens share a group."""
    is_img = input_ids == image_token_id
    prev = F.pad(is_img, (1, 0))[:, :-1]
    group = torch.cumsum((is_img & ~prev).int(), dim=1) - 1
    return torch.where(is_img, group, torch.full_like(group, -1))


class Gemma3ForConditionalGeneration(nn.Module):
    hf_architectures = ("Gemma3ForConditionalGeneration",)
    config_class = Gemma3VLMConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> Gemma3VLMConfig:
        return Gemma3VLMConfig.from_hf_config(hf_cfg)

    def __init__(self, config: Gemma3VLMConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = Gemma3VLMConfig(**config)
        self.config = config
        lm = Gemma3ForCausalLM(config.text, backend=backend)
        inner = nn.Module()
        inner.vision_tower = SiglipVisionTower(config.vision)
        inner.multi_modal_projector = Gemma3MultiModalProjector(config)
        inner.language_model = lm.model      # HF key layout
        self.model = inner
        self.lm_head = lm.lm_head
        self._lm = lm                        # forward logic holder (not in sd)
        # drop the duplicate module registration so state_dict keys are clean
        del self._modules["_lm"]
        object.__setattr__(self, "_lm", lm)
        self.loss_fn = None

    def forward(self, input_ids: torch.Tensor,
                pixel_values: torch.Tensor | None = None,
                labels: torch.Tensor | None = None, **_: Any) -> torch.Tensor:
        lm = self._lm
        lm.loss_fn = self.loss_fn
        embeds = self.model.language_model.embed_tokens(input_ids)
        block_ids = None
        if pixel_values is not None:
            feats = self.model.vision_tower(pixel_values)
            img = self.model.multi_modal_projector(feats) \
                .reshape(-1, embeds.shape[-1])
            mask = input_ids == self.config.image_token_id
            embeds = embeds.clone()
            embeds[mask] = img.to(embeds.dtype)
            block_ids = image_block_ids(input_ids, self.config.image_token_id)
        return lm(input_ids, labels=labels, inputs_embeds=embeds,
                  block_ids=block_ids)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
        self._lm.init_weights(device=None)
        std = 0.02
        for mod in self.model.vision_tower.modules():
            if isinstance(mod, (nn.Linear, nn.Conv2d, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, nn.LayerNorm):
                nn.init.ones_(mod.weight)
                nn.init.zeros_(mod.bias)
        proj = self.model.multi_modal_projector
        nn.init.normal_(proj.mm_input_projection_weight, std=std)
        nn.init.zeros_(proj.mm_soft_emb_norm.weight)
