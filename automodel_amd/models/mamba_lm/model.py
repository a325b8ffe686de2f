"""Pure state-space causal LMs: Mamba, Mamba2 (Codestral-style), FalconMamba.

Reference behavior: the public HF architectures (transformers.models.
{mamba, mamba2, falcon_mamba}) — backbone of residual pre-norm mixer
blocks (no attention, no FFN), final norm_f, tied lm_head (mamba/falcon).
Reuses the shared mixers: chunked-SSD Mamba2Mixer (models/nemotron_h) and
the chunked segsum Mamba-1 mixer (models/jamba) with dt/B/C norms off
(Mamba) or weightless (FalconMamba's distinctive stabilizer).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.jamba.model import JambaMambaMixer
from automodel_amd.models.nemotron_h.model import Mamba2Mixer
from automodel_amd.ops.rms_norm import RMSNorm


@dataclass
class MambaLMConfig:
    vocab_size: int = 50280
    hidden_size: int = 768
    num_hidden_layers: int = 24
    state_size: int = 16
    expand: int = 2
    conv_kernel: int = 4
    time_step_rank: int = 48            # mamba1/falcon
    num_heads: int = 128                # mamba2
    head_dim: int = 64                  # mamba2
    n_groups: int = 1                   # mamba2
    chunk_size: int = 256               # mamba2
    use_bias: bool = False
    use_conv_bias: bool = True
    layer_norm_epsilon: float = 1e-5
    mixer_rms_eps: float = 1e-6         # falcon weightless norms
    tie_word_embeddings: bool = True
    initializer_range: float = 0.1
    kind: str = "mamba"                 # mamba | mamba2 | falcon_mamba

    @classmethod
    def from_hf_config(cls, hf: Any, kind: str) -> "MambaLMConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        tsr = g("time_step_rank", "auto")
        if tsr == "auto":
            tsr = -(-g("hidden_size", 768) // 16)
        return cls(
            vocab_size=g("vocab_size", 50280),
            hidden_size=g("hidden_size", 768),
            num_hidden_layers=g("num_hidden_layers", 24),
            state_size=g("state_size", 16),
            expand=g("expand", 2),
            conv_kernel=g("conv_kernel", 4),
            time_step_rank=tsr,
            num_heads=g("num_heads", 128),
            head_dim=g("head_dim", 64),
            n_groups=g("n_groups", 1),
            chunk_size=g("chunk_size", 256),
            use_bias=g("use_bias", False),
            use_conv_bias=g("use_conv_bias", True),
            layer_norm_epsilon=g("layer_norm_epsilon", 1e-5),
            mixer_rms_eps=g("mixer_rms_eps", 1e-6),
            tie_word_embeddings=g("tie_word_embeddings", True),
            kind=kind,
        )


class _Block(nn.Module):
    def __init__(self, cfg: MambaLMConfig):
        super().__init__()
        self.norm = RMSNorm(cfg.hidden_size, cfg.layer_norm_epsilon, "torch")
        inter = cfg.expand * cfg.hidden_size
        if cfg.kind == "mamba2":
            self.mixer = Mamba2Mixer(
                cfg.hidden_size, cfg.num_heads, cfg.head_dim, cfg.state_size,
                cfg.n_groups, cfg.conv_kernel, cfg.chunk_size,
                cfg.layer_norm_epsilon, use_bias=cfg.use_bias,
                use_conv_bias=cfg.use_conv_bias,
                norm_group_size=inter // cfg.n_groups)
        else:
            self.mixer = JambaMambaMixer(
                cfg.hidden_size, inter, cfg.state_size, cfg.time_step_rank,
                cfg.conv_kernel, cfg.mixer_rms_eps, use_bias=cfg.use_bias,
                use_conv_bias=cfg.use_conv_bias,
                dt_bc_norms="weightless" if cfg.kind == "falcon_mamba" else "none")

    def forward(self, x):
        return x + self.mixer(self.norm(x))


class _Backbone(nn.Module):
    def __init__(self, cfg: MambaLMConfig):
        super().__init__()
        self.embeddings = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(_Block(cfg) for _ in range(cfg.num_hidden_layers))
        self.norm_f = RMSNorm(cfg.hidden_size, cfg.layer_norm_epsilon, "torch")

    def forward(self, ids):
        x = self.embeddings(ids)
        for layer in self.layers:
            x = layer(x)
        return self.norm_f(x)


class _MambaLMBase(nn.Module):
    config_class = MambaLMConfig
    kind = "mamba"

    @classmethod
    def config_from_hf(cls, hf_cfg) -> MambaLMConfig:
        return MambaLMConfig.from_hf_config(hf_cfg, cls.kind)

    def __init__(self, config: MambaLMConfig | dict, backend=None):
        super().__init__()
        cfg = (config if isinstance(config, MambaLMConfig)
               else MambaLMConfig(**{**dict(config), "kind": self.kind}))
        cfg.kind = self.kind
        self.config = cfg
        self.backbone = _Backbone(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.backbone.embeddings.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None,
                return_hidden=False, **_):
        h = self.backbone(input_ids)
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

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        cfg = self.config
        if device is not None:
            self.to_empty(device=device)
        std = cfg.initializer_range
        for m in self.modules():
            if isinstance(m, (nn.Linear, nn.Conv1d)):
                nn.init.normal_(m.weight, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif type(m).__name__ in ("RMSNorm", "GatedRMSNorm"):
                nn.init.ones_(m.weight)
            elif isinstance(m, Mamba2Mixer):
                nn.init.ones_(m.dt_bias)
                nn.init.zeros_(m.A_log)
                nn.init.ones_(m.D)
            elif isinstance(m, JambaMambaMixer):
                A = torch.arange(1, m.state_size + 1, dtype=torch.float32,
                                 device=m.A_log.device).expand(m.inter, -1)
                m.A_log.copy_(A.log())
                nn.init.ones_(m.D)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.backbone.embeddings.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())


class MambaForCausalLM(_MambaLMBase):
    hf_architectures = ("MambaForCausalLM",)
    kind = "mamba"


class Mamba2ForCausalLM(_MambaLMBase):
    hf_architectures = ("Mamba2ForCausalLM",)
    kind = "mamba2"


class FalconMambaForCausalLM(_MambaLMBase):
    hf_architectures = ("FalconMambaForCausalLM",)
    kind = "falcon_mamba"
