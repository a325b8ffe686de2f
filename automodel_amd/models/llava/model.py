"""LLaVA: CLIP-style vision tower + MLP projector + llama decoder.

Reference behavior: nemo_automodel/components/models/llava_onevision (and
the VLM recipe family). Implemented directly against the public LLaVA
architecture: CLIP ViT (class token + learned positions, pre-LN encoder
with quick-GELU MLP), features taken from ``vision_feature_layer`` (default
-2, class token dropped), a 2-layer GELU projector, and image-token splice
into the llama stack. State-dict keys match HF
LlavaForConditionalGeneration (parity-tested).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.llama.model import LlamaConfig, LlamaDecoderLayer
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import build_rope_cache


@dataclass
class ClipVisionConfig:
    hidden_size: int = 1024
    intermediate_size: int = 4096
    num_hidden_layers: int = 24
    num_attention_heads: int = 16
    image_size: int = 336
    patch_size: int = 14
    num_channels: int = 3
    layer_norm_eps: float = 1e-5

    @property
    def num_patches(self) -> int:
        return (self.image_size // self.patch_size) ** 2


@dataclass
class LlavaConfig:
    text: LlamaConfig = field(default_factory=LlamaConfig)
    vision: ClipVisionConfig = field(default_factory=ClipVisionConfig)
    image_token_id: int = 32000
    vision_feature_layer: int = -2
    vision_feature_select_strategy: str = "default"   # drop the class token

    def __post_init__(self):
        if isinstance(self.text, dict):
            self.text = LlamaConfig(**self.text)
        if isinstance(self.vision, dict):
            self.vision = ClipVisionConfig(**self.vision)

    @property
    def vocab_size(self):
        return self.text.vocab_size

    @property
    def num_hidden_layers(self):
        return self.text.num_hidden_layers

    @property
    def num_key_value_heads(self):
        return self.text.num_key_value_heads

    @property
    def head_dim(self):
        return self.text.head_dim

    @classmethod
    def from_hf_config(cls, hf: Any) -> "LlavaConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        if "text" in hf and "vision" in hf:   # our own exported config.json
            import dataclasses as _dc

            keys = {f.name for f in _dc.fields(cls)}
            return cls(**{k: v for k, v in hf.items() if k in keys})
        tc = hf.get("text_config", hf.get("text", {}))
        vc = hf.get("vision_config", hf.get("vision", {}))
        return cls(
            text=LlamaConfig.from_hf_config(
                dict(tc, architectures=[tc.get("model_type", "llama")])),
            vision=ClipVisionConfig(
                hidden_size=vc.get("hidden_size", 1024),
                intermediate_size=vc.get("intermediate_size", 4096),
                num_hidden_layers=vc.get("num_hidden_layers", 24),
                num_attention_heads=vc.get("num_attention_heads", 16),
                image_size=vc.get("image_size", 336),
                patch_size=vc.get("patch_size", 14),
                num_channels=vc.get("num_channels", 3),
                layer_norm_eps=vc.get("layer_norm_eps", 1e-5),
            ),
            image_token_id=hf.get("image_token_id",
                                  hf.get("image_token_index", 32000)),
            vision_feature_layer=hf.get("vision_feature_layer", -2),
            vision_feature_select_strategy=hf.get(
                "vision_feature_select_strategy", "default"),
        )


class ClipAttention(nn.Module):
    def __init__(self, cfg: ClipVisionConfig):
        super().__init__()
        H = cfg.num_attention_heads
        self.num_heads, self.head_dim = H, cfg.hidden_size // H
        self.q_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.k_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.v_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.out_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, N, _ = x.shape
        q = self.q_proj(x).view(B, N, self.num_heads, -1).transpose(1, 2)
        k = self.k_proj(x).view(B, N, self.num_heads, -1).transpose(1, 2)
        v = self.v_proj(x).view(B, N, self.num_heads, -1).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v)
        return self.out_proj(o.transpose(1, 2).reshape(B, N, -1))


class ClipEncoderLayer(nn.Module):
    def __init__(self, cfg: ClipVisionConfig):
        super().__init__()
        self.layer_norm1 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.layer_norm2 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.self_attn = ClipAttention(cfg)
        self.mlp = nn.Module()
        self.mlp.fc1 = nn.Linear(cfg.hidden_size, cfg.intermediate_size)
        self.mlp.fc2 = nn.Linear(cfg.intermediate_size, cfg.hidden_size)

    def forward(self, x):
        x = x + self.self_attn(self.layer_norm1(x))
        h = self.mlp.fc1(self.layer_norm2(x))
        h = h * torch.sigmoid(1.702 * h)          # quick-gelu
        return x + self.mlp.fc2(h)


class ClipVisionTower(nn.Module):
    """HF CLIPVisionModel layout (embeddings + pre_layrnorm [sic] + encoder
    + post_layernorm); returns ALL hidden states so the caller picks
    vision_feature_layer."""

    def __init__(self, cfg: ClipVisionConfig):
        super().__init__()
        self.cfg = cfg
        emb = nn.Module()
        emb.class_embedding = nn.Parameter(torch.zeros(cfg.hidden_size))
        emb.patch_embedding = nn.Conv2d(cfg.num_channels, cfg.hidden_size,
                                        kernel_size=cfg.patch_size,
                                        stride=cfg.patch_size, bias=False)
        emb.position_embedding = nn.Embedding(cfg.num_patches + 1, cfg.hidden_size)
        self.embeddings = emb
        self.pre_layrnorm = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        enc = nn.Module()
        enc.layers = nn.ModuleList(ClipEncoderLayer(cfg)
                                   for _ in range(cfg.num_hidden_layers))
        self.encoder = enc
        self.post_layernorm = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)

    def forward(self, pixel_values: torch.Tensor) -> list[torch.Tensor]:
        B = pixel_values.shape[0]
        patches = self.embeddings.patch_embedding(
            pixel_values.to(self.embeddings.patch_embedding.weight.dtype))
        patches = patches.flatten(2).transpose(1, 2)          # [B, P, H]
        cls = self.embeddings.class_embedding.expand(B, 1, -1)
        x = torch.cat([cls, patches], dim=1)
        x = x + self.embeddings.position_embedding.weight[None]
        x = self.pre_layrnorm(x)
        hiddens = [x]
        for layer in self.encoder.layers:
            x = layer(x)
            hiddens.append(x)
        return hiddens


class LlavaForConditionalGeneration(nn.Module):
    hf_architectures = ("LlavaForConditionalGeneration",)
    config_class = LlavaConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> LlavaConfig:
        return LlavaConfig.from_hf_config(hf_cfg)

    def __init__(self, config: LlavaConfig | dict,
                 backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = LlavaConfig(**config)
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend),
            device_type, head_dim=config.text.head_dim)
        self.config = config
        self.backend = backend
        tc = config.text
        inner = nn.Module()
        inner.vision_tower = ClipVisionTower(config.vision)
        proj = nn.Module()
        proj.linear_1 = nn.Linear(config.vision.hidden_size, tc.hidden_size)
        proj.linear_2 = nn.Linear(tc.hidden_size, tc.hidden_size)
        inner.multi_modal_projector = proj
        lm = nn.Module()
        lm.embed_tokens = nn.Embedding(tc.vocab_size, tc.hidden_size)
        lm.layers = nn.ModuleList(LlamaDecoderLayer(tc, backend)
                                  for _ in range(tc.num_hidden_layers))
        lm.norm = RMSNorm(tc.hidden_size, tc.rms_norm_eps, backend.rms_norm)
        cos, sin = build_rope_cache(tc.head_dim, tc.max_position_embeddings,
                                    tc.rope_theta, tc.rope_scaling)
        lm.register_buffer("rope_cos", cos, persistent=False)
        lm.register_buffer("rope_sin", sin, persistent=False)
        inner.language_model = lm
        self.model = inner
        self.lm_head = nn.Linear(tc.hidden_size, tc.vocab_size, bias=False)
        if tc.tie_word_embeddings:
            self.lm_head.weight = lm.embed_tokens.weight
        self.loss_fn = None

    def freeze_vision_tower(self) -> None:
        for p in self.model.vision_tower.parameters():
            p.requires_grad_(False)

    def image_features(self, pixel_values: torch.Tensor) -> torch.Tensor:
        hiddens = self.model.vision_tower(pixel_values)
        feats = hiddens[self.config.vision_feature_layer]
        if self.config.vision_feature_select_strategy == "default":
            feats = feats[:, 1:]                  # drop the class token
        p = self.model.multi_modal_projector
        return p.linear_2(F.gelu(p.linear_1(feats)))

    def forward(self, input_ids: torch.Tensor,
                pixel_values: torch.Tensor | None = None,
                labels: torch.Tensor | None = None,
                position_ids: torch.Tensor | None = None, **_: Any) -> torch.Tensor:
        lm = self.model.language_model
        x = lm.embed_tokens(input_ids)
        if pixel_values is not None:
            img = self.image_features(pixel_values).reshape(-1, x.shape[-1])
            mask = input_ids == self.config.image_token_id
            x = x.clone()
            x[mask] = img.to(x.dtype)
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = lm.rope_cos[:S], lm.rope_sin[:S]
        else:
            cos, sin = lm.rope_cos[position_ids[0]], lm.rope_sin[position_ids[0]]
        cos, sin = cos.float(), sin.float()
        for layer in lm.layers:
            x = layer(x, cos, sin)
        hidden = lm.norm(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            tc = self.config.text
            cos, sin = build_rope_cache(tc.head_dim, tc.max_position_embeddings,
                                        tc.rope_theta, tc.rope_scaling)
            lm = self.model.language_model
            lm.rope_cos.copy_(cos.to(lm.rope_cos.device))
            lm.rope_sin.copy_(sin.to(lm.rope_sin.device))
        std = 0.02
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding, nn.Conv2d)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, nn.LayerNorm):
                nn.init.ones_(mod.weight)
                nn.init.zeros_(mod.bias)
            elif isinstance(mod, RMSNorm):
                nn.init.ones_(mod.weight)
        nn.init.normal_(self.model.vision_tower.embeddings.class_embedding, std=std)
        if self.config.text.tie_word_embeddings:
            self.lm_head.weight = self.model.language_model.embed_tokens.weight
