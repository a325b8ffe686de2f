"""LLaVA-OneVision: SigLIP tower + anyres patching + qwen2 decoder.

Reference behavior: nemo_automodel's llava_onevision family
(components/models/llava_onevision/, ~911 LoC). Implemented directly
against the public LLaVA-OneVision architecture:

  * SigLIP ViT tower (no class token, learned positions, gelu-tanh MLP),
    features from ``vision_feature_layer`` (default -1, strategy "full");
  * anyres: each image ships as 1 base crop + an NxM grid of high-res
    crops chosen by ``select_best_resolution`` over ``image_grid_pinpoints``;
    grid features are re-assembled spatially, unpadded back to the true
    aspect ratio, optionally bilinear-downscaled to ``anyres_max_N``
    tokens, and an ``image_newline`` embedding column is appended per row;
  * videos: per-frame features bilinear-pooled to ceil(side/2), one
    newline token per video;
  * 2-layer GELU projector and token splice into the qwen2 text stack.

State-dict keys match HF LlavaOnevisionForConditionalGeneration
(parity-tested text / anyres-image / video paths).
"""

from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.gemma.vlm import SiglipEncoderLayer, SiglipVisionConfig
from automodel_amd.models.llama.model import LlamaConfig, LlamaDecoderLayer
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import build_rope_cache

_MODEL_TYPE_TO_ARCH = {
    "qwen2": "Qwen2ForCausalLM",
    "llama": "LlamaForCausalLM",
    "mistral": "MistralForCausalLM",
}


def select_best_resolution(original_size, possible_resolutions):
    """Pick the pinpoint maximizing effective resolution, then minimizing
    wasted area (public anyres algorithm)."""
    oh, ow = original_size
    best, best_eff, best_waste = None, 0, float("inf")
    for h, w in possible_resolutions:
        scale = min(w / ow, h / oh)
        dw, dh = int(ow * scale), int(oh * scale)
        eff = min(dw * dh, ow * oh)
        waste = h * w - eff
        if eff > best_eff or (eff == best_eff and waste < best_waste):
            best, best_eff, best_waste = (h, w), eff, waste
    return best


def unpad_image(t: torch.Tensor, original_size) -> torch.Tensor:
    """t [C, H, W]; crop the letterbox padding implied by original_size."""
    oh, ow = int(original_size[0]), int(original_size[1])
    ch, cw = t.shape[1:]
    if ow / oh > cw / ch:
        nh = int(round(oh * (cw / ow), 7))
        pad = (ch - nh) // 2
        return t[:, pad:ch - pad, :]
    nw = int(round(ow * (ch / oh), 7))
    pad = (cw - nw) // 2
    return t[:, :, pad:cw - pad]


@dataclass
class LlavaOnevisionConfig:
    text: LlamaConfig = field(default_factory=LlamaConfig)
    vision: SiglipVisionConfig = field(default_factory=SiglipVisionConfig)
    image_token_id: int = 151646
    video_token_id: int = 151647
    image_grid_pinpoints: list = field(default_factory=list)
    vision_feature_layer: int = -1
    vision_feature_select_strategy: str = "full"
    vision_aspect_ratio: str = "anyres_max_9"
    initializer_range: float = 0.02

    def __post_init__(self):
        if isinstance(self.text, dict):
            self.text = LlamaConfig(**self.text)
        if isinstance(self.vision, dict):
            self.vision = SiglipVisionConfig(**self.vision)

    @property
    def vocab_size(self):
        return self.text.vocab_size

    @property
    def num_hidden_layers(self):
        return self.text.num_hidden_layers

    @classmethod
    def from_hf_config(cls, hf: Any) -> "LlavaOnevisionConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        tc = dict(hf.get("text_config", {}))
        if not tc.get("architectures") and tc.get("model_type") in _MODEL_TYPE_TO_ARCH:
            tc["architectures"] = [_MODEL_TYPE_TO_ARCH[tc["model_type"]]]
        vc = hf.get("vision_config", {})
        return cls(
            text=LlamaConfig.from_hf_config(tc),
            vision=SiglipVisionConfig(
                hidden_size=vc.get("hidden_size", 1152),
                intermediate_size=vc.get("intermediate_size", 4304),
                num_hidden_layers=vc.get("num_hidden_layers", 26),
                num_attention_heads=vc.get("num_attention_heads", 14),
                image_size=vc.get("image_size", 384),
                patch_size=vc.get("patch_size", 14),
                num_channels=vc.get("num_channels", 3),
                layer_norm_eps=vc.get("layer_norm_eps", 1e-6),
            ),
            image_token_id=hf.get("image_token_id",
                                  hf.get("image_token_index", 151646)),
            video_token_id=hf.get("video_token_id",
                                  hf.get("video_token_index", 151647)),
            image_grid_pinpoints=list(hf.get("image_grid_pinpoints") or []),
            vision_feature_layer=hf.get("vision_feature_layer", -1),
            vision_feature_select_strategy=hf.get(
                "vision_feature_select_strategy", "full"),
            vision_aspect_ratio=hf.get("vision_aspect_ratio", "anyres_max_9"),
            initializer_range=hf.get("initializer_range", 0.02),
        )


class SiglipTowerNoHead(nn.Module):
    """SigLIP tower layout (flat, no pooling head); returns ALL hidden
    states so the caller picks the feature layer."""

    def __init__(self, cfg: SiglipVisionConfig):
        super().__init__()
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

    def forward(self, pixel_values: torch.Tensor) -> list[torch.Tensor]:
        vm = self
        x = vm.embeddings.patch_embedding(
            pixel_values.to(vm.embeddings.patch_embedding.weight.dtype))
        x = x.flatten(2).transpose(1, 2)
        x = x + vm.embeddings.position_embedding.weight[None]
        hiddens = [x]
        for layer in vm.encoder.layers:
            x = layer(x)
            hiddens.append(x)
        # HF records post_layernorm output as last_hidden_state only; the
        # hidden_states list (what feature_layer indexes) is pre-norm.
        return hiddens


class LlavaOnevisionForConditionalGeneration(nn.Module):
    hf_architectures = ("LlavaOnevisionForConditionalGeneration",)
    config_class = LlavaOnevisionConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> LlavaOnevisionConfig:
        return LlavaOnevisionConfig.from_hf_config(hf_cfg)

    def __init__(self, config: LlavaOnevisionConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = LlavaOnevisionConfig(**config)
        self.config = config
        tc = config.text
        backend = BackendConfig.resolve(
            backend, "cuda" if torch.cuda.is_available() else "cpu",
            head_dim=tc.head_dim)
        inner = nn.Module()
        inner.vision_tower = SiglipTowerNoHead(config.vision)
        proj = nn.Module()
        proj.linear_1 = nn.Linear(config.vision.hidden_size, tc.hidden_size)
        proj.linear_2 = nn.Linear(tc.hidden_size, tc.hidden_size)
        inner.multi_modal_projector = proj
        inner.image_newline = nn.Parameter(torch.zeros(tc.hidden_size))
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

    # ---- vision features --------------------------------------------------
    def _project(self, pixel_values: torch.Tensor) -> torch.Tensor:
        hiddens = self.model.vision_tower(pixel_values)
        feats = hiddens[self.config.vision_feature_layer]
        if self.config.vision_feature_select_strategy == "default":
            feats = feats[:, 1:]
        p = self.model.multi_modal_projector
        return p.linear_2(F.gelu(p.linear_1(feats)))

    def image_features(self, pixel_values: torch.Tensor,
                       image_sizes: torch.Tensor) -> torch.Tensor:
        cfg = self.config
        side = cfg.vision.image_size // cfg.vision.patch_size
        num_patches = []
        for hw in image_sizes.tolist():
            bh, bw = select_best_resolution(hw, cfg.image_grid_pinpoints)
            num_patches.append((bh // cfg.vision.image_size)
                               * (bw // cfg.vision.image_size) + 1)
        if pixel_values.dim() == 5:
            pixel_values = torch.cat(
                [pv[:n] for pv, n in zip(pixel_values, num_patches)], dim=0)
        feats = torch.split(self._project(pixel_values), num_patches, dim=0)
        out = []
        for i, f in enumerate(feats):
            if f.shape[0] > 1:
                base, grid = f[0], f[1:]
                gh, gw = select_best_resolution(image_sizes[i].tolist(),
                                                cfg.image_grid_pinpoints)
                nph, npw = gh // cfg.vision.image_size, gw // cfg.vision.image_size
                g = grid.view(nph, npw, side, side, -1)
                g = g.permute(4, 0, 2, 1, 3).contiguous().flatten(1, 2).flatten(2, 3)
                g = unpad_image(g, image_sizes[i].tolist())
                max_np = int(cfg.vision_aspect_ratio.removeprefix("anyres_max_"))
                c, ch, cw = g.shape
                ratio = math.sqrt(ch * cw / (max_np * side ** 2))
                if ratio > 1.1:
                    g = F.interpolate(g[None], [int(ch // ratio), int(cw // ratio)],
                                      mode="bilinear")[0]
                nl = self.model.image_newline[:, None, None] \
                    .expand(*g.shape[:-1], 1).to(g.dtype)
                g = torch.cat([g, nl], dim=-1).flatten(1, 2).transpose(0, 1)
                out.append(torch.cat([base, g], dim=0))
            else:
                out.append(torch.cat(
                    [f[0], self.model.image_newline[None].to(f.dtype)], dim=0))
        return torch.cat(out, dim=0)

    def video_features(self, pixel_values_videos: torch.Tensor) -> torch.Tensor:
        cfg = self.config
        B, T, C, H, W = pixel_values_videos.shape
        f = self._project(pixel_values_videos.reshape(B * T, C, H, W))
        side = cfg.vision.image_size // cfg.vision.patch_size
        f = f.view(B * T, side, side, -1).permute(0, 3, 1, 2)
        f = F.interpolate(f, size=[math.ceil(side / 2)] * 2, mode="bilinear")
        f = f.permute(0, 2, 3, 1).reshape(B, -1, f.shape[1])
        nl = self.model.image_newline[None, None].expand(B, 1, -1).to(f.dtype)
        return torch.cat([f, nl], dim=1).flatten(0, 1)

    # ---- forward ----------------------------------------------------------
    def forward(self, input_ids: torch.Tensor,
                pixel_values: torch.Tensor | None = None,
                image_sizes: torch.Tensor | None = None,
                pixel_values_videos: torch.Tensor | None = None,
                labels: torch.Tensor | None = None,
                position_ids: torch.Tensor | None = None, **_: Any):
        cfg = self.config
        lm = self.model.language_model
        x = lm.embed_tokens(input_ids)
        if pixel_values is not None:
            img = self.image_features(pixel_values, image_sizes)
            mask = input_ids == cfg.image_token_id
            assert int(mask.sum()) == img.shape[0], \
                f"image slots {int(mask.sum())} != features {img.shape[0]}"
            x = x.clone()
            x[mask] = img.to(x.dtype)
        if pixel_values_videos is not None:
            vid = self.video_features(pixel_values_videos)
            mask = input_ids == cfg.video_token_id
            assert int(mask.sum()) == vid.shape[0], \
                f"video slots {int(mask.sum())} != features {vid.shape[0]}"
            x = x.clone()
            x[mask] = vid.to(x.dtype)
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = lm.rope_cos[:S], lm.rope_sin[:S]
        else:
            cos, sin = lm.rope_cos[position_ids[0]], lm.rope_sin[position_ids[0]]
        cos, sin = cos.float(), sin.float()
        for layer in lm.layers:
            x = layer(x, cos, sin)
        hidden = lm.norm(x)
        if labels is not None and self.loss_fn is not None:
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        logits = self.lm_head(hidden)
        if labels is not None:
            return F.cross_entropy(
                logits.reshape(-1, logits.shape[-1]).float(), labels.reshape(-1),
                ignore_index=-100, reduction="sum")
        return logits

    def freeze_vision_tower(self) -> None:
        for p in self.model.vision_tower.parameters():
            p.requires_grad_(False)

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
        std = self.config.initializer_range
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
        nn.init.normal_(self.model.image_newline,
                        std=1 / math.sqrt(self.config.text.hidden_size))
        if self.config.text.tie_word_embeddings:
            self.lm_head.weight = self.model.language_model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
