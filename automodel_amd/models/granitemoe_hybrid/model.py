"""GraniteMoeHybrid (IBM Granite 4.0) causal LM, MI355X-native.

Reference behavior: the public GraniteMoeHybrid architecture (HF
transformers.models.granitemoehybrid) — per-layer Mamba2 OR attention mixer
(layer_types), granite multipliers (embedding/attention/residual/logits),
topk-then-softmax MoE routing with a dense shared MLP added to the routed
output, optional NoPE (position_embedding_type). Reuses the shared
chunked-SSD Mamba2Mixer (full-dim gated norm, as in Bamba) and the grouped
MoE machinery (moe/layers.py — stream-overlapped shared expert, device-side
group plans on GPU).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.nemotron_h.model import Mamba2Mixer
from automodel_amd.moe.config import MoEConfig
from automodel_amd.moe.layers import MoE
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope, build_rope_cache
from automodel_amd.ops.swiglu import swiglu


@dataclass
class GraniteMoeHybridConfig:
    vocab_size: int = 32000
    hidden_size: int = 1536
    intermediate_size: int = 512          # routed-expert intermediate
    shared_intermediate_size: int = 1024
    num_hidden_layers: int = 40
    num_attention_heads: int = 12
    num_key_value_heads: int = 4
    num_local_experts: int = 62
    num_experts_per_tok: int = 6
    router_aux_loss_coef: float = 0.0
    layer_types: list = field(default_factory=list)   # "mamba"/"attention"
    mamba_n_heads: int = 48
    mamba_d_head: int = 64
    mamba_d_state: int = 128
    mamba_n_groups: int = 1
    mamba_conv_bias: bool = True
    mamba_proj_bias: bool = False
    mamba_chunk_size: int = 256
    conv_kernel: int = 4
    attention_bias: bool = False
    rms_norm_eps: float = 1e-5
    rope_theta: float = 10000.0
    rope_scaling: dict | None = None
    position_embedding_type: str = "nope"
    embedding_multiplier: float = 1.0
    attention_multiplier: float = 1.0
    residual_multiplier: float = 1.0
    logits_scaling: float = 1.0
    max_position_embeddings: int = 131072
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @classmethod
    def from_hf_config(cls, hf: Any) -> "GraniteMoeHybridConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        layer_types = g("layers_block_type") or g("layer_types") or []
        layer_types = ["mamba" if t in ("mamba", "linear_attention") else "attention"
                       for t in layer_types]
        arch = (g("architectures") or [""])[0]
        if arch == "GraniteMoeSharedForCausalLM":
            # all-attention granite with a shared MLP; rope always on
            layer_types = ["attention"] * g("num_hidden_layers", 32)
        return cls(
            vocab_size=g("vocab_size", 32000),
            hidden_size=g("hidden_size", 1536),
            intermediate_size=g("intermediate_size", 512),
            shared_intermediate_size=g("shared_intermediate_size", 1024),
            num_hidden_layers=g("num_hidden_layers", len(layer_types) or 40),
            num_attention_heads=g("num_attention_heads", 12),
            num_key_value_heads=g("num_key_value_heads", 4),
            num_local_experts=g("num_local_experts", 62),
            num_experts_per_tok=g("num_experts_per_tok", 6),
            router_aux_loss_coef=g("router_aux_loss_coef", 0.0),
            layer_types=layer_types,
            mamba_n_heads=g("mamba_n_heads", 48),
            mamba_d_head=g("mamba_d_head", 64),
            mamba_d_state=g("mamba_d_state", 128),
            mamba_n_groups=g("mamba_n_groups", 1),
            mamba_conv_bias=g("mamba_conv_bias", True),
            mamba_proj_bias=g("mamba_proj_bias", False),
            mamba_chunk_size=g("mamba_chunk_size", 256),
            conv_kernel=g("mamba_d_conv", 4),
            attention_bias=g("attention_bias", False),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            rope_scaling=g("rope_scaling"),
            position_embedding_type=("rope" if arch == "GraniteMoeSharedForCausalLM"
                                     else g("position_embedding_type") or "nope"),
            embedding_multiplier=g("embedding_multiplier", 1.0),
            attention_multiplier=g("attention_multiplier", 1.0),
            residual_multiplier=g("residual_multiplier", 1.0),
            logits_scaling=g("logits_scaling", 1.0),
            max_position_embeddings=g("max_position_embeddings", 131072),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class GraniteHybridAttention(nn.Module):
    """GQA with granite attention_multiplier as the softmax scale; rope only
    when position_embedding_type == "rope" (NoPE otherwise)."""

    def __init__(self, cfg: GraniteMoeHybridConfig, backend: BackendConfig):
        super().__init__()
        H, Hk = cfg.num_attention_heads, cfg.num_key_value_heads
        D = cfg.hidden_size // H
        self.head_dim = D
        self.scale = cfg.attention_multiplier
        self.use_rope = cfg.position_embedding_type == "rope"
        b = cfg.attention_bias
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=b)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=b)
        self.backend = backend

    def forward(self, h, cos, sin):
        B, S, _ = h.shape
        D = self.head_dim
        q = self.q_proj(h).view(B, S, -1, D)
        k = self.k_proj(h).view(B, S, -1, D)
        v = self.v_proj(h).view(B, S, -1, D)
        if self.use_rope:
            q, k = apply_rope(q, k, cos, sin, backend=self.backend.rope)
        o = flash_attention(q, k, v, causal=True, scale=self.scale,
                            backend=self.backend.attn)
        return self.o_proj(o.reshape(B, S, -1))


class GraniteSharedMLP(nn.Module):
    """Dense shared MLP with fused gate|up input_linear (HF key layout)."""

    def __init__(self, cfg: GraniteMoeHybridConfig):
        super().__init__()
        self.input_linear = nn.Linear(cfg.hidden_size,
                                      2 * cfg.shared_intermediate_size, bias=False)
        self.output_linear = nn.Linear(cfg.shared_intermediate_size,
                                       cfg.hidden_size, bias=False)

    def forward(self, x):
        gate, up = self.input_linear(x).chunk(2, dim=-1)
        return self.output_linear(swiglu(gate, up))


class GraniteMoeHybridLayer(nn.Module):
    def __init__(self, cfg: GraniteMoeHybridConfig, backend: BackendConfig,
                 layer_idx: int):
        super().__init__()
        types = cfg.layer_types or ["mamba"] * cfg.num_hidden_layers
        self.is_attn = types[layer_idx] == "attention"
        self.residual_multiplier = cfg.residual_multiplier
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                                backend.rms_norm)
        if self.is_attn:
            self.self_attn = GraniteHybridAttention(cfg, backend)
        else:
            self.mamba = Mamba2Mixer(
                cfg.hidden_size, cfg.mamba_n_heads, cfg.mamba_d_head,
                cfg.mamba_d_state, cfg.mamba_n_groups, cfg.conv_kernel,
                cfg.mamba_chunk_size, cfg.rms_norm_eps,
                use_bias=cfg.mamba_proj_bias, use_conv_bias=cfg.mamba_conv_bias,
                norm_group_size=None)   # full-dim gated norm (Bamba lineage)
        self.block_sparse_moe = None
        if cfg.num_local_experts > 0:
            self.block_sparse_moe = MoE(cfg.hidden_size, MoEConfig(
                n_routed_experts=cfg.num_local_experts,
                n_activated_experts=cfg.num_experts_per_tok,
                moe_intermediate_size=cfg.intermediate_size,
                aux_loss_coeff=cfg.router_aux_loss_coef,
                topk_then_softmax=True))
        self.shared_mlp = GraniteSharedMLP(cfg)

    def forward(self, x, cos, sin):
        r = self.residual_multiplier
        h = self.input_layernorm(x)
        h = self.self_attn(h, cos, sin) if self.is_attn else self.mamba(h)
        x = x + h * r
        h = self.post_attention_layernorm(x)
        y = self.shared_mlp(h)
        if self.block_sparse_moe is not None:
            y = y + self.block_sparse_moe(h)
        return x + y * r


class GraniteMoeHybridModel(nn.Module):
    def __init__(self, cfg: GraniteMoeHybridConfig, backend: BackendConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            GraniteMoeHybridLayer(cfg, backend, i)
            for i in range(cfg.num_hidden_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        D = cfg.hidden_size // cfg.num_attention_heads
        cos, sin = build_rope_cache(D, min(cfg.max_position_embeddings, 32768),
                                    cfg.rope_theta, cfg.rope_scaling)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, ids):
        x = self.embed_tokens(ids) * self.cfg.embedding_multiplier
        S = x.shape[1]
        cos, sin = self.rope_cos[:S].float(), self.rope_sin[:S].float()
        for layer in self.layers:
            x = layer(x, cos, sin)
        return self.norm(x)


class GraniteMoeHybridStateDictAdapter:
    """HF <-> in-tree key/layout mapping.

    HF: block_sparse_moe.router.weight, experts.gate_up_proj [E,2I,H],
        experts.down_proj [E,H,I]
    mine: block_sparse_moe.gate.weight, experts.{gate,up}_proj [E,I,H],
          experts.down_proj [E,H,I]
    """

    def from_hf(self, sd: dict) -> dict:
        out = {}
        for k, v in sd.items():
            if k.endswith("block_sparse_moe.router.weight"):
                out[k.replace(".router.weight", ".gate.weight")] = v
            elif k.endswith("experts.gate_up_proj"):
                gate, up = v.chunk(2, dim=1)
                out[k.replace("gate_up_proj", "gate_proj")] = gate.contiguous()
                out[k.replace("gate_up_proj", "up_proj")] = up.contiguous()
            else:
                out[k] = v
        return out

    def to_hf(self, sd: dict) -> dict:
        out = {}
        for k, v in sd.items():
            if k.endswith("block_sparse_moe.gate.weight"):
                out[k.replace(".gate.weight", ".router.weight")] = v
            elif k.endswith("experts.gate_proj"):
                up = sd[k.replace("gate_proj", "up_proj")]
                out[k.replace("gate_proj", "gate_up_proj")] = torch.cat([v, up], dim=1)
            elif k.endswith("experts.up_proj"):
                continue
            else:
                out[k] = v
        return out


class GraniteMoeHybridForCausalLM(nn.Module):
    hf_architectures = ("GraniteMoeHybridForCausalLM",
                        "GraniteMoeSharedForCausalLM")
    config_class = GraniteMoeHybridConfig
    state_dict_adapter = GraniteMoeHybridStateDictAdapter

    @staticmethod
    def config_from_hf(hf_cfg) -> GraniteMoeHybridConfig:
        return GraniteMoeHybridConfig.from_hf_config(hf_cfg)

    def __init__(self, config: GraniteMoeHybridConfig | dict, backend=None):
        super().__init__()
        cfg = (config if isinstance(config, GraniteMoeHybridConfig)
               else GraniteMoeHybridConfig(**dict(config)))
        self.config = cfg
        bk = BackendConfig.resolve(backend,
                                   "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.hidden_size // cfg.num_attention_heads)
        self.model = GraniteMoeHybridModel(cfg, bk)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None,
                return_hidden=False, **_):
        h = self.model(input_ids)
        if return_hidden:
            return h
        if labels is not None and self.loss_fn is not None:
            return self.loss_fn(h, self.lm_head.weight, labels)
        logits = self.lm_head(h) / self.config.logits_scaling
        if labels is not None:
            return torch.nn.functional.cross_entropy(
                logits.reshape(-1, logits.shape[-1]).float(), labels.reshape(-1),
                ignore_index=-100, reduction="sum")
        return logits

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        cfg = self.config
        if device is not None:
            self.to_empty(device=device)
            D = cfg.hidden_size // cfg.num_attention_heads
            cos, sin = build_rope_cache(D, min(cfg.max_position_embeddings, 32768),
                                        cfg.rope_theta, cfg.rope_scaling, device=device)
            self.model.rope_cos.copy_(cos)
            self.model.rope_sin.copy_(sin)
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
            elif isinstance(m, MoE):
                nn.init.normal_(m.gate.weight, std=std)
                nn.init.normal_(m.experts.gate_proj, std=std)
                nn.init.normal_(m.experts.up_proj, std=std)
                nn.init.normal_(m.experts.down_proj, std=std)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
