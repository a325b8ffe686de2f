"""Llama-4 text causal LM (Scout/Maverick-class), MI355X-native.

Reference behavior: the public Llama4 text architecture (HF
transformers.models.llama4) — the NeMo reference covers it through its
generic transformers path; here it is a first-class custom model:
  * interleaved MoE/dense decoder layers (interleave_moe_layer_step)
  * router: top-k on LOGITS, per-replica sigmoid score applied to the
    expert INPUT (not the output combine) + always-on shared expert
  * NoPE layers (no_rope_layers) with attention temperature tuning
    (log1p(floor(pos/floor_scale)) * attn_scale + 1 on queries)
  * interleaved-pair rope (complex convention) — handled by the same
    de-interleave permutation the llama family uses (score-invariant)
  * weightless L2 qk-norm AFTER rope on rope layers (use_qk_norm)
Experts run the in-tree grouped-GEMM kernels; attention runs the HIP flash
kernels via ops/attention.py.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.moe.experts import GroupedExperts, permute_tokens, unpermute_tokens
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope, build_rope_cache
from automodel_amd.ops.swiglu import swiglu


@dataclass
class Llama4Config:
    vocab_size: int = 202048
    hidden_size: int = 5120
    intermediate_size: int = 8192        # expert dim
    intermediate_size_mlp: int = 16384   # dense/shared MLP dim
    num_hidden_layers: int = 48
    num_attention_heads: int = 40
    num_key_value_heads: int = 8
    head_dim: int = 128
    num_local_experts: int = 16
    num_experts_per_tok: int = 1
    interleave_moe_layer_step: int = 1
    no_rope_layers: list | None = None   # 1 = rope, 0 = NoPE (HF convention)
    use_qk_norm: bool = True
    attn_temperature_tuning: bool = True
    attn_scale: float = 0.1
    floor_scale: float = 8192.0
    attention_chunk_size: int | None = 8192
    rope_theta: float = 500000.0
    rope_scaling: dict | None = None
    rms_norm_eps: float = 1e-5
    max_position_embeddings: int = 8192
    attention_bias: bool = False
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    def __post_init__(self):
        if self.no_rope_layers is None:
            self.no_rope_layers = [
                0 if (i + 1) % 4 == 0 else 1 for i in range(self.num_hidden_layers)
            ]

    def is_moe_layer(self, i: int) -> bool:
        return (i + 1) % self.interleave_moe_layer_step == 0

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Llama4Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        hf = hf.get("text_config", hf)
        g = hf.get
        rp = g("rope_parameters") or {}
        theta = rp.get("rope_theta", g("rope_theta", 500000.0))
        scaling = g("rope_scaling")
        if scaling is None and rp.get("rope_type", "default") != "default":
            scaling = rp
        return cls(
            vocab_size=g("vocab_size", 202048),
            hidden_size=g("hidden_size", 5120),
            intermediate_size=g("intermediate_size", 8192),
            intermediate_size_mlp=g("intermediate_size_mlp", 16384),
            num_hidden_layers=g("num_hidden_layers", 48),
            num_attention_heads=g("num_attention_heads", 40),
            num_key_value_heads=g("num_key_value_heads", 8),
            head_dim=g("head_dim", 128),
            num_local_experts=g("num_local_experts", 16),
            num_experts_per_tok=g("num_experts_per_tok", 1),
            interleave_moe_layer_step=g("interleave_moe_layer_step", 1),
            no_rope_layers=g("no_rope_layers"),
            use_qk_norm=g("use_qk_norm", True),
            attn_temperature_tuning=bool(g("attn_temperature_tuning", True)),
            attn_scale=g("attn_scale", 0.1),
            floor_scale=g("floor_scale", 8192.0),
            attention_chunk_size=g("attention_chunk_size", 8192),
            rope_theta=theta,
            rope_scaling=scaling,
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            max_position_embeddings=g("max_position_embeddings", 8192),
            attention_bias=g("attention_bias", False),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


def _l2norm(x: torch.Tensor, eps: float) -> torch.Tensor:
    return x * torch.rsqrt(x.float().pow(2).mean(-1, keepdim=True) + eps).to(x.dtype)


class Llama4Attention(nn.Module):
    def __init__(self, cfg: Llama4Config, backend: BackendConfig, layer_idx: int):
        super().__init__()
        self.cfg = cfg
        self.backend = backend
        self.layer_idx = layer_idx
        self.use_rope = bool(cfg.no_rope_layers[layer_idx])
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        b = cfg.attention_bias
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=b)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=b)

    def forward(self, x, cos, sin):
        cfg = self.cfg
        B, S, _ = x.shape
        D = cfg.head_dim
        q = self.q_proj(x).view(B, S, -1, D)
        k = self.k_proj(x).view(B, S, -1, D)
        v = self.v_proj(x).view(B, S, -1, D)
        if self.use_rope:
            # complex/interleaved rope: de-interleave both q and k to the
            # half-split convention (dot products are permutation-invariant)
            d2 = D // 2
            q = q.reshape(B, S, -1, d2, 2).transpose(-1, -2).reshape(B, S, -1, D)
            k = k.reshape(B, S, -1, d2, 2).transpose(-1, -2).reshape(B, S, -1, D)
            q, k = apply_rope(q.contiguous(), k.contiguous(), cos, sin,
                              backend=self.backend.rope)
            if cfg.use_qk_norm:
                q = _l2norm(q, cfg.rms_norm_eps)
                k = _l2norm(k, cfg.rms_norm_eps)
        elif cfg.attn_temperature_tuning:
            pos = torch.arange(S, device=x.device).float()
            scales = (torch.log1p(torch.floor((pos + 1.0) / cfg.floor_scale))
                      * cfg.attn_scale + 1.0)
            q = (q * scales.view(1, S, 1, 1)).to(q.dtype)
        o = flash_attention(q, k, v, causal=True, backend=self.backend.attn)
        return self.o_proj(o.reshape(B, S, -1))


class Llama4MLP(nn.Module):
    def __init__(self, hidden: int, inter: int):
        super().__init__()
        self.gate_proj = nn.Linear(hidden, inter, bias=False)
        self.up_proj = nn.Linear(hidden, inter, bias=False)
        self.down_proj = nn.Linear(inter, hidden, bias=False)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


class Llama4MoE(nn.Module):
    """Router top-k on logits, sigmoid scores applied to the expert INPUT,
    plus an always-on shared expert (the HF Llama4TextMoe semantics)."""

    def __init__(self, cfg: Llama4Config, backend: BackendConfig):
        super().__init__()
        E = cfg.num_local_experts
        self.top_k = cfg.num_experts_per_tok
        self.n_experts = E
        self.router = nn.Linear(cfg.hidden_size, E, bias=False)
        self.experts = GroupedExperts(E, cfg.hidden_size, cfg.intermediate_size,
                                      backend=backend.experts)
        self.shared_expert = Llama4MLP(cfg.hidden_size, cfg.intermediate_size)
        self.last_expert_load: torch.Tensor | None = None

    def forward(self, x):
        B, S, H = x.shape
        xf = x.reshape(-1, H)
        logits = self.router(xf)
        top_v, top_i = logits.topk(self.top_k, dim=-1)
        scores = torch.sigmoid(top_v.float()).to(x.dtype)        # [T, k]
        with torch.no_grad():
            load = torch.zeros(self.n_experts, device=x.device)
            load.scatter_add_(0, top_i.reshape(-1),
                              torch.ones(top_i.numel(), device=x.device))
            self.last_expert_load = load
        x_perm, sort_idx, counts = permute_tokens(xf, top_i, self.n_experts)
        perm_scores = scores.reshape(-1)[sort_idx]
        y_perm = self.experts.forward_permuted(x_perm * perm_scores.unsqueeze(1),
                                               counts)
        ones = torch.ones_like(scores)
        y = unpermute_tokens(y_perm, sort_idx, ones)
        return (y + self.shared_expert(xf)).view(B, S, H)


class Llama4DecoderLayer(nn.Module):
    def __init__(self, cfg: Llama4Config, backend: BackendConfig, layer_idx: int):
        super().__init__()
        self.self_attn = Llama4Attention(cfg, backend, layer_idx)
        if cfg.is_moe_layer(layer_idx):
            self.feed_forward = Llama4MoE(cfg, backend)
        else:
            self.feed_forward = Llama4MLP(cfg.hidden_size, cfg.intermediate_size_mlp)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                                backend.rms_norm)

    def forward(self, x, cos, sin):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin)
        x = x + self.feed_forward(self.post_attention_layernorm(x))
        return x


class Llama4Model(nn.Module):
    def __init__(self, cfg: Llama4Config, backend: BackendConfig):
        super().__init__()
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            Llama4DecoderLayer(cfg, backend, i) for i in range(cfg.num_hidden_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        cos, sin = build_rope_cache(cfg.head_dim, cfg.max_position_embeddings,
                                    cfg.rope_theta, cfg.rope_scaling)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, ids):
        x = self.embed_tokens(ids)
        S = x.shape[1]
        cos, sin = self.rope_cos[:S].float(), self.rope_sin[:S].float()
        for layer in self.layers:
            x = layer(x, cos, sin)
        return self.norm(x)


class Llama4StateDictAdapter:
    """HF bmm-layout expert weights <-> our stacked grouped layout:
    gate_up_proj [E, H, 2I] -> gate/up [E, I, H]; down [E, I, H] -> [E, H, I]."""

    def from_hf(self, sd: dict) -> dict:
        out = {}
        for k, t in sd.items():
            if k.endswith(".feed_forward.experts.gate_up_proj"):
                inter = t.shape[2] // 2
                out[k.replace("gate_up_proj", "gate_proj")] = \
                    t[..., :inter].transpose(1, 2).contiguous()
                out[k.replace("gate_up_proj", "up_proj")] = \
                    t[..., inter:].transpose(1, 2).contiguous()
            elif k.endswith(".feed_forward.experts.down_proj"):
                out[k] = t.transpose(1, 2).contiguous()
            else:
                out[k] = t
        return out

    def to_hf(self, sd: dict) -> dict:
        out = {}
        for k, t in sd.items():
            if k.endswith(".feed_forward.experts.gate_proj"):
                up = sd[k.replace("gate_proj", "up_proj")]
                out[k.replace("gate_proj", "gate_up_proj")] = torch.cat(
                    [t.transpose(1, 2), up.transpose(1, 2)], dim=-1).contiguous()
            elif k.endswith(".feed_forward.experts.up_proj"):
                continue
            elif k.endswith(".feed_forward.experts.down_proj"):
                out[k] = t.transpose(1, 2).contiguous()
            else:
                out[k] = t
        return out

    def hf_key_targets(self, key: str) -> list:
        if key.endswith(".feed_forward.experts.gate_up_proj"):
            return [key.replace("gate_up_proj", "gate_proj"),
                    key.replace("gate_up_proj", "up_proj")]
        return [key]


class Llama4ForCausalLM(nn.Module):
    hf_architectures = ("Llama4ForCausalLM", "Llama4ForConditionalGeneration")
    config_class = Llama4Config

    @staticmethod
    def config_from_hf(hf_cfg) -> Llama4Config:
        return Llama4Config.from_hf_config(hf_cfg)

    def __init__(self, config: Llama4Config | dict, backend=None):
        super().__init__()
        cfg = config if isinstance(config, Llama4Config) else Llama4Config(**dict(config))
        self.config = cfg
        bk = BackendConfig.resolve(backend, "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.head_dim)
        self.model = Llama4Model(cfg, bk)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.state_dict_adapter = Llama4StateDictAdapter()
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None,
                return_hidden=False, **_):
        h = self.model(input_ids)
        if return_hidden:
            return h
        if labels is not None and self.loss_fn is not None:
            return self.loss_fn(h, self.lm_head.weight, labels)
        logits = self.lm_head(h)
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
            cos, sin = build_rope_cache(cfg.head_dim, cfg.max_position_embeddings,
                                        cfg.rope_theta, cfg.rope_scaling, device=device)
            self.model.rope_cos.copy_(cos)
            self.model.rope_sin.copy_(sin)
        std = cfg.initializer_range
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, RMSNorm):
                nn.init.ones_(m.weight)
            elif isinstance(m, GroupedExperts):
                m.init_weights(std)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        seen, total = set(), 0
        for p in self.parameters():
            if id(p) not in seen:
                seen.add(id(p))
                total += p.numel()
        return total
