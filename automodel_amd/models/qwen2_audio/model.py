"""Qwen2-Audio (Whisper-style audio encoder + LLM), MI355X-native.

Reference behavior: the public Qwen2-Audio architecture (HF
transformers.models.qwen2_audio) — mel features through conv1(gelu) +
stride-2 conv2(gelu), learned positions, pre-LN bidirectional encoder
(k_proj bias-free), stride-2 average pooling, final LayerNorm, a linear
projector, and audio embeddings spliced at audio_token_id positions of the
text model (reuses the in-tree llama/qwen2 stack). Adds the AUDIO modality
to the multimodal coverage (text+vision stacks elsewhere).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM

_MODEL_TYPE_TO_ARCH = {
    "qwen2": "Qwen2ForCausalLM",
    "llama": "LlamaForCausalLM",
    "mistral": "MistralForCausalLM",
}


@dataclass
class Qwen2AudioEncoderConfig:
    d_model: int = 1280
    encoder_layers: int = 32
    encoder_attention_heads: int = 20
    encoder_ffn_dim: int = 5120
    num_mel_bins: int = 128
    max_source_positions: int = 1500
    activation_function: str = "gelu"


@dataclass
class Qwen2AudioConfig:
    text: dict = field(default_factory=dict)
    audio: Qwen2AudioEncoderConfig = field(default_factory=Qwen2AudioEncoderConfig)
    audio_token_id: int = 151646
    initializer_range: float = 0.02

    def __post_init__(self):
        if isinstance(self.audio, dict):
            self.audio = Qwen2AudioEncoderConfig(**self.audio)

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Qwen2AudioConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        t = dict(hf.get("text_config", {}))
        if not t.get("architectures") and t.get("model_type") in _MODEL_TYPE_TO_ARCH:
            t["architectures"] = [_MODEL_TYPE_TO_ARCH[t["model_type"]]]
        a = hf.get("audio_config", {})
        audio = Qwen2AudioEncoderConfig(
            d_model=a.get("d_model", 1280),
            encoder_layers=a.get("encoder_layers", 32),
            encoder_attention_heads=a.get("encoder_attention_heads", 20),
            encoder_ffn_dim=a.get("encoder_ffn_dim", 5120),
            num_mel_bins=a.get("num_mel_bins", 128),
            max_source_positions=a.get("max_source_positions", 1500),
            activation_function=a.get("activation_function", "gelu"),
        )
        return cls(text=t, audio=audio,
                   audio_token_id=hf.get("audio_token_id") or hf.get("audio_token_index", 151646))


class AudioEncoderLayer(nn.Module):
    """Pre-LN Whisper encoder layer (bidirectional)."""

    def __init__(self, cfg: Qwen2AudioEncoderConfig):
        super().__init__()
        D = cfg.d_model
        self.n_heads = cfg.encoder_attention_heads
        attn = nn.Module()
        attn.q_proj = nn.Linear(D, D, bias=True)
        attn.k_proj = nn.Linear(D, D, bias=False)
        attn.v_proj = nn.Linear(D, D, bias=True)
        attn.out_proj = nn.Linear(D, D, bias=True)
        self.self_attn = attn
        self.self_attn_layer_norm = nn.LayerNorm(D)
        self.fc1 = nn.Linear(D, cfg.encoder_ffn_dim)
        self.fc2 = nn.Linear(cfg.encoder_ffn_dim, D)
        self.final_layer_norm = nn.LayerNorm(D)
        self.act = F.gelu if cfg.activation_function == "gelu" else getattr(F, cfg.activation_function)

    def forward(self, x):
        B, S, D = x.shape
        a = self.self_attn
        h = self.self_attn_layer_norm(x)
        q = a.q_proj(h).view(B, S, self.n_heads, -1).transpose(1, 2)
        k = a.k_proj(h).view(B, S, self.n_heads, -1).transpose(1, 2)
        v = a.v_proj(h).view(B, S, self.n_heads, -1).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v)
        x = x + a.out_proj(o.transpose(1, 2).reshape(B, S, D))
        return x + self.fc2(self.act(self.fc1(self.final_layer_norm(x))))


class Qwen2AudioEncoder(nn.Module):
    def __init__(self, cfg: Qwen2AudioEncoderConfig):
        super().__init__()
        D = cfg.d_model
        self.conv1 = nn.Conv1d(cfg.num_mel_bins, D, kernel_size=3, padding=1)
        self.conv2 = nn.Conv1d(D, D, kernel_size=3, stride=2, padding=1)
        self.embed_positions = nn.Embedding(cfg.max_source_positions, D)
        self.embed_positions.requires_grad_(False)
        self.layers = nn.ModuleList(AudioEncoderLayer(cfg)
                                    for _ in range(cfg.encoder_layers))
        self.layer_norm = nn.LayerNorm(D)
        self.avg_pooler = nn.AvgPool1d(2, stride=2)

    def forward(self, input_features: torch.Tensor) -> torch.Tensor:
        x = F.gelu(self.conv1(input_features))
        x = F.gelu(self.conv2(x)).permute(0, 2, 1)
        x = x + self.embed_positions.weight[: x.shape[1]]
        for layer in self.layers:
            x = layer(x)
        x = self.avg_pooler(x.permute(0, 2, 1)).permute(0, 2, 1)
        return self.layer_norm(x)


class Qwen2AudioForConditionalGeneration(nn.Module):
    hf_architectures = ("Qwen2AudioForConditionalGeneration",)
    config_class = Qwen2AudioConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> Qwen2AudioConfig:
        return Qwen2AudioConfig.from_hf_config(hf_cfg)

    def __init__(self, config: Qwen2AudioConfig | dict, backend=None):
        super().__init__()
        cfg = (config if isinstance(config, Qwen2AudioConfig)
               else Qwen2AudioConfig(**dict(config)))
        self.config = cfg
        inner = nn.Module()
        inner.audio_tower = Qwen2AudioEncoder(cfg.audio)
        proj = nn.Module()
        tcfg = LlamaConfig.from_hf_config(dict(cfg.text))
        proj.linear = nn.Linear(cfg.audio.d_model, tcfg.hidden_size, bias=True)
        inner.multi_modal_projector = proj
        lm = LlamaForCausalLM(tcfg, backend=backend)
        inner.language_model = lm.model      # share HF's nesting (model.language_model)
        self.model = inner
        # keep the full LM accessible WITHOUT registering duplicate params
        self.__dict__["_lm_holder"] = [lm]
        self.lm_head = lm.lm_head
        self.loss_fn = None

    def forward(self, input_ids, input_features=None, labels=None,
                position_ids=None, return_hidden=False, **_):
        cfg = self.config
        lm = self._lm_holder[0]
        embeds = self.model.language_model.embed_tokens(input_ids)
        if input_features is not None and input_features.numel() > 0:
            audio = self.model.audio_tower(input_features.to(embeds.dtype))
            audio = self.model.multi_modal_projector.linear(audio)
            mask = input_ids == cfg.audio_token_id
            assert int(mask.sum()) == audio.shape[0] * audio.shape[1], \
                "audio slots != encoder tokens"
            embeds = embeds.clone()
            embeds[mask] = audio.reshape(-1, audio.shape[-1]).to(embeds.dtype)
        h = lm.forward_embeds(embeds, position_ids=position_ids) \
            if hasattr(lm, "forward_embeds") else self._decode(embeds, position_ids)
        if return_hidden:
            return h
        if labels is not None and self.loss_fn is not None:
            return self.loss_fn(h, self.lm_head.weight, labels)
        logits = self.lm_head(h)
        if labels is not None:
            return F.cross_entropy(
                logits.reshape(-1, logits.shape[-1]).float(), labels.reshape(-1),
                ignore_index=-100, reduction="sum")
        return logits

    def _decode(self, embeds, position_ids):
        m = self.model.language_model
        S = embeds.shape[1]
        if position_ids is not None:
            cos = m.rope_cos[position_ids[0]].float()
            sin = m.rope_sin[position_ids[0]].float()
        else:
            cos, sin = m.rope_cos[:S].float(), m.rope_sin[:S].float()
        x = embeds
        for layer in m.layers:
            x = layer(x, cos, sin)
        return m.norm(x)

    def freeze_audio_tower(self) -> None:
        for p in self.model.audio_tower.parameters():
            p.requires_grad_(False)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
        self._lm_holder[0].init_weights(device=device)
        std = self.config.initializer_range
        for m in (*self.model.audio_tower.modules(),
                  *self.model.multi_modal_projector.modules()):
            if isinstance(m, (nn.Linear, nn.Conv1d)):
                nn.init.normal_(m.weight, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, nn.LayerNorm):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
