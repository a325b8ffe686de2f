"""HunYuan-MoE-v1 causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
Llama-shaped RMS pre-norm blocks with per-head qk RMSNorm applied AFTER
rope, a softmax→top-k→renorm MoE over STACKED expert tensors
(gate_up [E,2I,H] / down [E,H,I]) with an always-on ``shared_mlp``, and
an fp32 router ``gate.wg``. HF keys match HunYuanMoEV1ForCausalLM
(parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope_ref, build_rope_cache


@dataclass
class HunYuanMoEV1Config:
    vocab_size: int = 129024
    hidden_size: int = 4096
    intermediate_size: int = 3072
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    head_dim: int = 128
    num_experts: int = 16
    moe_topk: int = 2
    num_shared_expert: int = 1
    attention_bias: bool = False
    max_position_embeddings: int = 32768
    rope_theta: float = 10000.0
    rms_norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @classmethod
    def from_hf_config(cls, hf: Any) -> "HunYuanMoEV1Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get

        def scalar(v, default):
            if v is None:
                return default
            return v[0] if isinstance(v, (list, tuple)) else v

        return cls(
            vocab_size=g("vocab_size", 129024),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 3072),
            num_hidden_layers=g("num_hidden_layers", 32),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads", 8),
            head_dim=g("head_dim") or g("hidden_size", 4096) // g("num_attention_heads", 32),
            num_experts=scalar(g("num_experts"), 16),
            moe_topk=scalar(g("moe_topk"), 2),
            num_shared_expert=scalar(g("num_shared_expert"), 1),
            attention_bias=g("attention_bias", False),
            max_position_embeddings=g("max_position_embeddings", 32768),
            rope_theta=(g("rope_parameters") or {}).get(
                "rope_theta", g("rope_theta", 10000.0)),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class HunYuanLayer(nn.Module):
    def __init__(self, cfg: HunYuanMoEV1Config):
        super().__init__()
        H, Hk, D, E = (cfg.num_attention_heads, cfg.num_key_value_heads,
                       cfg.head_dim, cfg.hidden_size)
        self.H, self.Hk, self.D = H, Hk, D
        self.top_k, self.n_exp = cfg.moe_topk, cfg.num_experts
        self.input_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(E, eps=cfg.rms_norm_eps)
        b = cfg.attention_bias
        attn = nn.Module()
        attn.q_proj = nn.Linear(E, H * D, bias=b)
        attn.k_proj = nn.Linear(E, Hk * D, bias=b)
        attn.v_proj = nn.Linear(E, Hk * D, bias=b)
        attn.o_proj = nn.Linear(H * D, E, bias=b)
        attn.query_layernorm = RMSNorm(D, eps=cfg.rms_norm_eps)
        attn.key_layernorm = RMSNorm(D, eps=cfg.rms_norm_eps)
        self.self_attn = attn
        I = cfg.intermediate_size
        mlp = nn.Module()
        gate = nn.Module()
        gate.wg = nn.Linear(E, cfg.num_experts, bias=False)
        mlp.gate = gate
        experts = nn.Module()
        experts.gate_up_proj = nn.Parameter(torch.empty(cfg.num_experts, 2 * I, E))
        experts.down_proj = nn.Parameter(torch.empty(cfg.num_experts, E, I))
        mlp.experts = experts
        shared = nn.Module()
        si = I * cfg.num_shared_expert
        shared.gate_proj = nn.Linear(E, si, bias=False)
        shared.up_proj = nn.Linear(E, si, bias=False)
        shared.down_proj = nn.Linear(si, E, bias=False)
        mlp.shared_mlp = shared
        self.mlp = mlp

    def _moe(self, x):
        m = self.mlp
        B, S, E = x.shape
        shared = m.shared_mlp.down_proj(
            F.silu(m.shared_mlp.gate_proj(x)) * m.shared_mlp.up_proj(x))
        xf = x.reshape(-1, E)
        probs = F.softmax(F.linear(xf.float(), m.gate.wg.weight.float()), dim=-1)
        weights, idx = torch.topk(probs, self.top_k, dim=-1)
        weights = (weights / weights.sum(dim=-1, keepdim=True)).to(x.dtype)
        out = torch.zeros_like(xf)
        for e in idx.unique():
            tok, slot = torch.where(idx == e)
            gate, up = F.linear(xf[tok], m.experts.gate_up_proj[e]).chunk(2, dim=-1)
            h = F.silu(gate) * up
            out.index_add_(0, tok, F.linear(h, m.experts.down_proj[e])
                           * weights[tok, slot, None])
        return out.view(B, S, E) + shared

    def forward(self, x, cos, sin):
        B, S, _ = x.shape
        h = self.input_layernorm(x)
        a = self.self_attn
        q = a.q_proj(h).view(B, S, self.H, self.D)
        k = a.k_proj(h).view(B, S, self.Hk, self.D)
        v = a.v_proj(h).view(B, S, self.Hk, self.D)
        q, k = apply_rope_ref(q, k, cos, sin)
        q = a.query_layernorm(q)        # qk-norm AFTER rope
        k = a.key_layernorm(k)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, enable_gqa=self.H != self.Hk)
        x = x + a.o_proj(o.transpose(1, 2).reshape(B, S, -1))
        return x + self._moe(self.post_attention_layernorm(x))


class HunYuanMoEV1ForCausalLM(nn.Module):
    hf_architectures = ("HunYuanMoEV1ForCausalLM",)
    config_class = HunYuanMoEV1Config

    @staticmethod
    def config_from_hf(hf_cfg) -> HunYuanMoEV1Config:
        return HunYuanMoEV1Config.from_hf_config(hf_cfg)

    def __init__(self, config: HunYuanMoEV1Config | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = HunYuanMoEV1Config(**config)
        self.config = config
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(HunYuanLayer(config)
                                     for _ in range(config.num_hidden_layers))
        inner.norm = RMSNorm(config.hidden_size, eps=config.rms_norm_eps)
        cos, sin = build_rope_cache(config.head_dim,
                                    config.max_position_embeddings,
                                    config.rope_theta)
        inner.register_buffer("rope_cos", cos, persistent=False)
        inner.register_buffer("rope_sin", sin, persistent=False)
        self.model = inner
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = inner.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None, **_: Any):
        m = self.model
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = m.rope_cos[:S], m.rope_sin[:S]
        else:
            cos, sin = m.rope_cos[position_ids[0]], m.rope_sin[position_ids[0]]
        cos, sin = cos.float(), sin.float()
        x = m.embed_tokens(input_ids)
        for layer in m.layers:
            x = layer(x, cos, sin)
        hidden = m.norm(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            cos, sin = build_rope_cache(self.config.head_dim,
                                        self.config.max_position_embeddings,
                                        self.config.rope_theta)
            self.model.rope_cos.copy_(cos.to(self.model.rope_cos.device))
            self.model.rope_sin.copy_(sin.to(self.model.rope_sin.device))
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, RMSNorm):
                nn.init.ones_(mod.weight)
        for layer in self.model.layers:
            nn.init.normal_(layer.mlp.experts.gate_up_proj, std=std)
            nn.init.normal_(layer.mlp.experts.down_proj, std=std)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
