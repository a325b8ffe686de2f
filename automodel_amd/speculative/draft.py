"""EAGLE-style draft model for speculative decoding.

Reference behavior: nemo_automodel/components/speculative/eagle/
draft_llama.py:652 (Eagle3LlamaModel: embed_tokens + fc projecting the
concatenated target aux hidden states to draft width + fused first decoder
layer over [embed, hidden] 2H features + norm + lm_head) and core.py's TTT
training. This is an independent, smaller design with the same contract:

  * ``fc``: num_aux * H_target -> H maps the target's 3 auxiliary hidden
    states into the draft width (EAGLE-3's multi-level feature fusion);
  * ``fuse``: 2H -> H combines the current token's embedding with the
    carried hidden state (EAGLE's [embed, hidden] fused input);
  * a stack of standard LlamaDecoderLayers (default 1) — the draft reuses
    this framework's HIP flash-attention/rms/rope/swiglu kernels as-is;
  * lm_head tied to the target's (frozen) so draft logits live in the
    target vocab without extra memory.

At decode time the draft chains its own last hidden state into positions
the target has not verified yet (EAGLE's hidden-state recycling).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.llama.model import LlamaConfig, LlamaDecoderLayer
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import build_rope_cache


@dataclass
class EagleDraftConfig:
    hidden_size: int = 4096            # draft width (== target width usually)
    target_hidden_size: int | None = None
    num_aux_hidden_states: int = 3
    num_layers: int = 1
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    intermediate_size: int = 14336
    vocab_size: int = 128256
    rope_theta: float = 500000.0
    max_position_embeddings: int = 8192
    rms_norm_eps: float = 1e-5

    @classmethod
    def from_target(cls, target_cfg: LlamaConfig, **overrides) -> "EagleDraftConfig":
        d = dict(
            hidden_size=target_cfg.hidden_size,
            target_hidden_size=target_cfg.hidden_size,
            num_attention_heads=target_cfg.num_attention_heads,
            num_key_value_heads=target_cfg.num_key_value_heads,
            intermediate_size=target_cfg.intermediate_size,
            vocab_size=target_cfg.vocab_size,
            rope_theta=target_cfg.rope_theta,
            max_position_embeddings=target_cfg.max_position_embeddings,
            rms_norm_eps=target_cfg.rms_norm_eps,
        )
        d.update(overrides)
        return cls(**d)


class EagleDraftModel(nn.Module):
    def __init__(self, cfg: EagleDraftConfig, backend: BackendConfig | None = None):
        super().__init__()
        self.cfg = cfg
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(backend, device_type)
        H = cfg.hidden_size
        Ht = cfg.target_hidden_size or H
        layer_cfg = LlamaConfig(
            vocab_size=cfg.vocab_size, hidden_size=H,
            intermediate_size=cfg.intermediate_size,
            num_hidden_layers=cfg.num_layers,
            num_attention_heads=cfg.num_attention_heads,
            num_key_value_heads=cfg.num_key_value_heads,
            rope_theta=cfg.rope_theta,
            max_position_embeddings=cfg.max_position_embeddings,
            rms_norm_eps=cfg.rms_norm_eps,
        )
        self.embed_tokens = nn.Embedding(cfg.vocab_size, H)
        self.fc = nn.Linear(Ht * cfg.num_aux_hidden_states, H, bias=False)
        self.fuse = nn.Linear(2 * H, H, bias=False)
        self.layers = nn.ModuleList(
            LlamaDecoderLayer(layer_cfg, backend) for _ in range(cfg.num_layers)
        )
        self.norm = RMSNorm(H, cfg.rms_norm_eps, backend.rms_norm)
        self.lm_head = nn.Linear(H, cfg.vocab_size, bias=False)
        cos, sin = build_rope_cache(layer_cfg.head_dim, cfg.max_position_embeddings,
                                    cfg.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def tie_to_target(self, target) -> None:
        """Share (frozen) embed + lm_head with the target CausalLM."""
        self.embed_tokens.weight = target.model.embed_tokens.weight
        self.lm_head.weight = target.lm_head.weight
        self.embed_tokens.weight.requires_grad_(False)
        self.lm_head.weight.requires_grad_(False)

    def fuse_aux(self, aux: list[torch.Tensor]) -> torch.Tensor:
        """[B,S,Ht] x num_aux -> [B,S,H] draft hidden carrier."""
        assert len(aux) == self.cfg.num_aux_hidden_states, len(aux)
        return self.fc(torch.cat(aux, dim=-1))

    def backbone(self, input_ids: torch.Tensor, carry: torch.Tensor) -> torch.Tensor:
        """-> draft hidden [B,S,H]. carry[t] is the hidden paired with token t
        (the target's fused aux at verified positions, the draft's own
        previous hidden at speculative positions)."""
        x = self.fuse(torch.cat([self.embed_tokens(input_ids), carry], dim=-1))
        S = input_ids.shape[1]
        cos, sin = self.rope_cos[:S], self.rope_sin[:S]
        if cos.dtype != torch.float32:
            cos, sin = cos.float(), sin.float()
        for layer in self.layers:
            x = layer(x, cos, sin)
        return self.norm(x)

    def forward(self, input_ids: torch.Tensor, carry: torch.Tensor) -> torch.Tensor:
        return self.lm_head(self.backbone(input_ids, carry))


def load_draft(path: str, target, device="cpu") -> "EagleDraftModel":
    """Load a trained EAGLE draft from a checkpoint dir (draft.pt +
    draft_config.json written by TrainEagleDraftRecipe / torch.save)."""
    import json
    import os

    import torch

    cfg_path = os.path.join(path, "draft_config.json")
    if os.path.exists(cfg_path):
        with open(cfg_path) as f:
            cfg = EagleDraftConfig(**json.load(f))
    else:
        cfg = EagleDraftConfig.from_target(target.config)
    draft = EagleDraftModel(cfg)
    sd_path = os.path.join(path, "draft.pt")
    if os.path.exists(sd_path):
        draft.load_state_dict(torch.load(sd_path, map_location="cpu",
                                         weights_only=True), strict=False)
    draft.tie_to_target(target)
    return draft.to(device)
