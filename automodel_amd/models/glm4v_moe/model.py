"""GLM-4.5V (Glm4vMoe): GLM-4V EVA-style ViT + GLM4-MoE text on 3D MRoPE.

Reference behavior: nemo_automodel's glm4_moe VLM lineage (components/
models/glm4_moe/ + the glm4v vision stack). Implemented directly against
the public GLM-4.5V architecture:

  * vision tower: identical to GLM-4V (grid_sample bicubic pos resample,
    RMS norms, SwiGLU vision MLPs, merge-conv downsample) — reused;
  * text: GLM4-MoE decoder layers (DeepSeek-style sigmoid routing with
    e_score_correction_bias, group top-k, shared expert, dense-first
    layers, PLAIN pre/post norms — no GLM-4 sandwich norms) with PARTIAL
    rotary (factor 0.5, contiguous halves) driven by CHUNKED-section 3D
    MRoPE tables;
  * image splice + get_rope_index shared with GLM-4V.

HF keys match Glm4vMoeForConditionalGeneration via the fused-expert
adapter (experts.gate_up_proj split), parity-tested text and image paths.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.glm4_moe.model import Glm4MoeConfig, Glm4MoeDecoderLayer
from automodel_amd.models.glm4v.model import (
    Glm4vConfig,
    Glm4vForConditionalGeneration,
    Glm4vVisionConfig,
    Glm4vVisionModel,
)
from automodel_amd.moe.layers import MoE
from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter
from automodel_amd.ops.rms_norm import RMSNorm


@dataclass
class Glm4vMoeConfig:
    text: Glm4MoeConfig = field(default_factory=Glm4MoeConfig)
    vision: Glm4vVisionConfig = field(default_factory=Glm4vVisionConfig)
    mrope_section: tuple = (8, 12, 12)
    image_token_id: int = 151363
    video_token_id: int = 151364
    initializer_range: float = 0.02

    def __post_init__(self):
        if isinstance(self.text, dict):
            self.text = Glm4MoeConfig(**self.text)
        if isinstance(self.vision, dict):
            self.vision = Glm4vVisionConfig(**self.vision)
        self.mrope_section = tuple(self.mrope_section)

    @property
    def vocab_size(self):
        return self.text.vocab_size

    @property
    def num_hidden_layers(self):
        return self.text.num_hidden_layers

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Glm4vMoeConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        tc = dict(hf.get("text_config", {}))
        rp = tc.get("rope_parameters") or tc.get("rope_scaling") or {}
        tc.setdefault("rope_theta", rp.get("rope_theta", 10000.0))
        tc.setdefault("partial_rotary_factor",
                      rp.get("partial_rotary_factor", 0.5))
        base = Glm4vConfig.from_hf_config(hf)   # parses the shared vision cfg
        return cls(
            text=Glm4MoeConfig.from_hf_config(tc),
            vision=base.vision,
            mrope_section=tuple(rp.get("mrope_section", (8, 12, 12))),
            image_token_id=hf.get("image_token_id", 151363),
            video_token_id=hf.get("video_token_id", 151364),
            initializer_range=hf.get("initializer_range", 0.02),
        )


class Glm4vMoeTextModel(nn.Module):
    """GLM4-MoE decoder stack under chunked-section 3D MRoPE tables."""

    def __init__(self, cfg: Glm4MoeConfig, mrope_section: tuple,
                 backend: BackendConfig):
        super().__init__()
        self.cfg = cfg
        self.mrope_section = mrope_section
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            Glm4MoeDecoderLayer(cfg, backend, i)
            for i in range(cfg.num_hidden_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        rot = int(cfg.head_dim * cfg.partial_rotary_factor)
        inv = 1.0 / (cfg.rope_theta ** (torch.arange(0, rot, 2).float() / rot))
        self.register_buffer("inv_freq", inv, persistent=False)

    def _mrope(self, position_ids: torch.Tensor, dtype):
        freqs = torch.einsum("nbs,d->nbsd", position_ids.float(),
                             self.inv_freq.float())
        chunks = freqs.split(list(self.mrope_section), dim=-1)
        out = torch.cat([c[i % 3] for i, c in enumerate(chunks)], dim=-1)
        emb = torch.cat([out, out], dim=-1)
        return emb.cos().to(dtype), emb.sin().to(dtype)

    def forward(self, embeds, position_ids):
        cos, sin = self._mrope(position_ids, torch.float32)
        x = embeds
        for layer in self.layers:
            x = layer(x, cos, sin)
        return self.norm(x)


class Glm4vMoeForConditionalGeneration(Glm4vForConditionalGeneration):
    hf_architectures = ("Glm4vMoeForConditionalGeneration",)
    config_class = Glm4vMoeConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> Glm4vMoeConfig:
        return Glm4vMoeConfig.from_hf_config(hf_cfg)

    def __init__(self, config: Glm4vMoeConfig | dict, backend=None):
        nn.Module.__init__(self)
        cfg = (config if isinstance(config, Glm4vMoeConfig)
               else Glm4vMoeConfig(**dict(config)))
        self.config = cfg
        bk = BackendConfig.resolve(backend,
                                   "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.text.head_dim)
        self.state_dict_adapter = MoEStateDictAdapter(cfg.text)
        inner = nn.Module()
        inner.visual = Glm4vVisionModel(cfg.vision)
        inner.language_model = Glm4vMoeTextModel(cfg.text, cfg.mrope_section, bk)
        self.model = inner
        self.lm_head = nn.Linear(cfg.text.hidden_size, cfg.text.vocab_size,
                                 bias=False)
        if cfg.text.tie_word_embeddings:
            self.lm_head.weight = inner.language_model.embed_tokens.weight
        self.loss_fn = None

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        super().init_weights(device=device)
        std = self.config.initializer_range
        for m in self.modules():
            if isinstance(m, MoE):
                m.experts.init_weights(std)
                nn.init.normal_(m.gate.weight, std=std)
                if getattr(m.gate, "e_score_correction_bias", None) is not None:
                    m.gate.e_score_correction_bias.zero_()
