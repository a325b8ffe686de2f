"""Vision-language model: ViT tower -> MLP projector -> Llama-family LLM.

Reference behavior: nemo_automodel/components/models/qwen3_vl (and the VLM
recipe recipes/vlm/finetune.py): a vision tower encodes images to patch
embeddings, a projector maps them into the LLM hidden size, and image
placeholder tokens in the text sequence are replaced by the projected
embeddings before the decoder stack.

The vision tower runs torch-SDPA attention (non-causal, head_dim != 128);
the LLM tower runs the full HIP kernel path. Tower freezing follows the
reference's frozen-tower handling (finetune.py:108-125).
"""

from __future__ import annotations

from dataclasses import dataclass, field

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM


@dataclass
class VisionConfig:
    image_size: int = 224
    patch_size: int = 14
    hidden_size: int = 256
    intermediate_size: int = 1024
    num_hidden_layers: int = 4
    num_attention_heads: int = 4
    layer_norm_eps: float = 1e-6
    num_channels: int = 3

    @property
    def num_patches(self) -> int:
        return (self.image_size // self.patch_size) ** 2


@dataclass
class VLMConfig:
    text: LlamaConfig = field(default_factory=LlamaConfig)
    vision: VisionConfig = field(default_factory=VisionConfig)
    image_token_id: int = 151655   # Qwen2-VL convention

    def __post_init__(self):
        if isinstance(self.text, dict):
            self.text = LlamaConfig(**self.text)
        if isinstance(self.vision, dict):
            self.vision = VisionConfig(**self.vision)


class VisionBlock(nn.Module):
    def __init__(self, cfg: VisionConfig):
        super().__init__()
        self.norm1 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.attn = nn.MultiheadAttention(cfg.hidden_size, cfg.num_attention_heads,
                                          batch_first=True)
        self.norm2 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.mlp = nn.Sequential(
            nn.Linear(cfg.hidden_size, cfg.intermediate_size),
            nn.GELU(),
            nn.Linear(cfg.intermediate_size, cfg.hidden_size),
        )

    def forward(self, x):
        h = self.norm1(x)
        x = x + self.attn(h, h, h, need_weights=False)[0]
        return x + self.mlp(self.norm2(x))


class VisionTower(nn.Module):
    def __init__(self, cfg: VisionConfig):
        super().__init__()
        self.cfg = cfg
        self.patch_embed = nn.Conv2d(cfg.num_channels, cfg.hidden_size,
                                     kernel_size=cfg.patch_size, stride=cfg.patch_size)
        self.pos_embed = nn.Parameter(torch.zeros(1, cfg.num_patches, cfg.hidden_size))
        self.blocks = nn.ModuleList(VisionBlock(cfg) for _ in range(cfg.num_hidden_layers))
        self.norm = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)

    def forward(self, pixel_values: torch.Tensor) -> torch.Tensor:
        """pixel_values [N_img, C, H, W] -> [N_img, num_patches, hidden]."""
        x = self.patch_embed(pixel_values).flatten(2).transpose(1, 2)
        x = x + self.pos_embed
        for blk in self.blocks:
            x = blk(x)
        return self.norm(x)


class VLMForConditionalGeneration(nn.Module):
    hf_architectures = ("GenericVLMForConditionalGeneration",)  # qwen2_vl/ has the real Qwen2-VL
    config_class = VLMConfig

    def __init__(self, config: VLMConfig | dict, backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = VLMConfig(**config)
        self.config = config
        self.visual = VisionTower(config.vision)
        self.projector = nn.Sequential(
            nn.Linear(config.vision.hidden_size, config.text.hidden_size),
            nn.GELU(),
            nn.Linear(config.text.hidden_size, config.text.hidden_size),
        )
        self.language_model = LlamaForCausalLM(config.text, backend=backend)
        self.loss_fn = None

    def freeze_vision_tower(self) -> None:
        for p in self.visual.parameters():
            p.requires_grad_(False)

    def forward(self, input_ids, pixel_values=None, labels=None, position_ids=None,
                **_):
        embeds = self.language_model.model.embed_tokens(input_ids)
        if pixel_values is not None and pixel_values.numel() > 0:
            img = self.projector(self.visual(pixel_values.to(embeds.dtype)))
            img = img.reshape(-1, img.shape[-1])  # [N_img*P, H]
            mask = input_ids == self.config.image_token_id
            n_slots = int(mask.sum())
            assert n_slots == img.shape[0], (
                f"image token slots ({n_slots}) != patch embeds ({img.shape[0]})"
            )
            embeds = embeds.clone()
            embeds[mask] = img.to(embeds.dtype)
        # run the decoder on the mixed embeddings
        lm = self.language_model
        x = embeds
        S = x.shape[1]
        cos, sin = lm.model.rope_cos[:S], lm.model.rope_sin[:S]
        if cos.dtype != torch.float32:
            cos, sin = cos.float(), sin.float()
        if position_ids is not None:
            cos, sin = lm.model.rope_cos[position_ids[0]].float(), lm.model.rope_sin[position_ids[0]].float()
        for layer in lm.model.layers:
            x = layer(x, cos, sin)
        x = lm.model.norm(x)
        if labels is not None:
            assert self.loss_fn is not None
            return self.loss_fn(x, lm.lm_head.weight, labels)
        return lm.lm_head(x)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
        self.language_model.init_weights(device=device)
        for m in self.visual.modules():
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
        for m in self.projector.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=0.02)
                nn.init.zeros_(m.bias)
