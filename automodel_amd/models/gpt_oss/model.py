"""GPT-OSS-family causal LM (MoE + attention sinks), MI355X-native.

Reference behavior: nemo_automodel/components/models/gpt_oss/ (MoE decoder
with sink attention). Architecture deltas vs the llama/qwen MoE families:

  * attention SINKS: a learned per-head logit joins the softmax and its
    probability mass is dropped — heads can "attend to nothing";
  * q/k/v/o projections carry biases; sliding window on alternating layers
    (config.layer_types);
  * experts store HF layout [E, H, 2I] with INTERLEAVED gate/up columns and
    biases; activation is the clamped glu  (up+1) * gate*sigmoid(1.702*gate)
    with gate clamped at +7, up at +-7;
  * router: top-k FIRST, softmax over the k selected logits;
  * YaRN rope (build_rope_cache "yarn" branch) at 128k context.

head_dim 64 routes attention to sdpa via BackendConfig.resolve; the expert
loop gathers tokens per hit expert (same structure as moe/experts.py's eager
path). State-dict keys match HF GptOssForCausalLM exactly.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope, build_rope_cache


@dataclass
class GptOssConfig:
    vocab_size: int = 201088
    hidden_size: int = 2880
    intermediate_size: int = 2880
    num_hidden_layers: int = 24
    num_attention_heads: int = 64
    num_key_value_heads: int = 8
    head_dim: int = 64
    num_local_experts: int = 32
    num_experts_per_tok: int = 4
    sliding_window: int = 128
    layer_types: list | None = None       # default: alternate sliding/full
    max_position_embeddings: int = 131072
    rope_theta: float = 150000.0
    rope_scaling: dict | None = None      # yarn for the released checkpoints
    rms_norm_eps: float = 1e-5
    attention_bias: bool = True
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    def __post_init__(self):
        if self.layer_types is None:
            self.layer_types = [
                "sliding_attention" if i % 2 == 0 else "full_attention"
                for i in range(self.num_hidden_layers)
            ]

    @classmethod
    def from_hf_config(cls, hf: Any) -> "GptOssConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rope = g("rope_parameters") or g("rope_scaling")
        theta = (rope or {}).get("rope_theta", g("rope_theta", 150000.0))
        if rope and rope.get("rope_type", "default") == "default":
            rope = None  # plain rotary; theta already extracted above
        return cls(
            vocab_size=g("vocab_size", 201088),
            hidden_size=g("hidden_size", 2880),
            intermediate_size=g("intermediate_size", 2880),
            num_hidden_layers=g("num_hidden_layers", 24),
            num_attention_heads=g("num_attention_heads", 64),
            num_key_value_heads=g("num_key_value_heads", 8),
            head_dim=g("head_dim", 64),
            num_local_experts=g("num_local_experts", 32),
            num_experts_per_tok=g("num_experts_per_tok", 4),
            sliding_window=g("sliding_window", 128),
            layer_types=g("layer_types"),
            max_position_embeddings=g("max_position_embeddings", 131072),
            rope_theta=theta,
            rope_scaling=rope,
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            attention_bias=g("attention_bias", True),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class GptOssAttention(nn.Module):
    def __init__(self, cfg: GptOssConfig, backend: BackendConfig, layer_idx: int):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.num_heads, self.num_kv_heads, self.head_dim = H, Hk, D
        b = cfg.attention_bias
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=b)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=b)
        self.sinks = nn.Parameter(torch.zeros(H))
        self.scale = D**-0.5
        self.window = (cfg.sliding_window
                       if cfg.layer_types[layer_idx] == "sliding_attention" else None)
        self.backend = backend

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        B, S, _ = x.shape
        q = self.q_proj(x).view(B, S, -1, self.head_dim)
        k = self.k_proj(x).view(B, S, -1, self.head_dim)
        v = self.v_proj(x).view(B, S, -1, self.head_dim)
        q, k = apply_rope(q, k, cos, sin, backend=self.backend.rope)
        # sink attention: append one learned logit column per head to the
        # softmax, drop its probability (reference gpt_oss sink semantics)
        qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))
        rep = self.num_heads // self.num_kv_heads
        if rep > 1:
            kt = kt.repeat_interleave(rep, dim=1)
            vt = vt.repeat_interleave(rep, dim=1)
        scores = torch.matmul(qt.float(), kt.float().transpose(-1, -2)) * self.scale
        i = torch.arange(S, device=x.device)
        mask = i[None, :] > i[:, None]
        if self.window is not None:
            mask |= i[None, :] <= (i[:, None] - self.window)
        scores = scores.masked_fill(mask, float("-inf"))
        sinks = self.sinks.float().reshape(1, -1, 1, 1).expand(B, -1, S, 1)
        combined = torch.cat([scores, sinks], dim=-1)
        combined = combined - combined.amax(dim=-1, keepdim=True)
        probs = combined.softmax(-1)[..., :-1]           # drop the sink mass
        o = torch.matmul(probs, vt.float()).to(x.dtype).transpose(1, 2)
        return self.o_proj(o.reshape(B, S, -1))


class GptOssExperts(nn.Module):
    """HF-layout stacked experts ([E, H, 2I] interleaved gate/up, biases)."""

    alpha = 1.702
    limit = 7.0

    def __init__(self, cfg: GptOssConfig):
        super().__init__()
        E, Hd, I = cfg.num_local_experts, cfg.hidden_size, cfg.intermediate_size
        self.num_experts = E
        self.gate_up_proj = nn.Parameter(torch.empty(E, Hd, 2 * I))
        self.gate_up_proj_bias = nn.Parameter(torch.empty(E, 2 * I))
        self.down_proj = nn.Parameter(torch.empty(E, I, Hd))
        self.down_proj_bias = nn.Parameter(torch.empty(E, Hd))

    def act(self, gate_up: torch.Tensor) -> torch.Tensor:
        gate, up = gate_up[..., ::2], gate_up[..., 1::2]
        gate = gate.clamp(max=self.limit)
        up = up.clamp(-self.limit, self.limit)
        return (up + 1) * (gate * torch.sigmoid(gate * self.alpha))

    def forward(self, x: torch.Tensor, indices: torch.Tensor,
                weights: torch.Tensor) -> torch.Tensor:
        """x [T, H]; indices/weights [T, k] -> [T, H]."""
        out = torch.zeros_like(x)
        for e in range(self.num_experts):
            slot_tok, slot_k = torch.where(indices == e)
            if slot_tok.numel() == 0:
                continue
            xe = x[slot_tok]
            h = self.act(xe @ self.gate_up_proj[e] + self.gate_up_proj_bias[e])
            ye = h @ self.down_proj[e] + self.down_proj_bias[e]
            out.index_add_(0, slot_tok, ye * weights[slot_tok, slot_k, None])
        return out


class GptOssRouter(nn.Module):
    """top-k FIRST, then softmax over the selected logits."""

    def __init__(self, cfg: GptOssConfig):
        super().__init__()
        self.top_k = cfg.num_experts_per_tok
        self.weight = nn.Parameter(torch.zeros(cfg.num_local_experts, cfg.hidden_size))
        self.bias = nn.Parameter(torch.zeros(cfg.num_local_experts))

    def forward(self, x: torch.Tensor):
        logits = F.linear(x, self.weight, self.bias)
        top_vals, top_idx = torch.topk(logits, self.top_k, dim=-1)
        return top_vals.softmax(-1), top_idx


class GptOssMLP(nn.Module):
    def __init__(self, cfg: GptOssConfig):
        super().__init__()
        self.router = GptOssRouter(cfg)
        self.experts = GptOssExperts(cfg)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, Hd = x.shape
        flat = x.reshape(-1, Hd)
        weights, idx = self.router(flat)
        return self.experts(flat, idx, weights).reshape(B, S, Hd)


class GptOssDecoderLayer(nn.Module):
    def __init__(self, cfg: GptOssConfig, backend: BackendConfig, layer_idx: int):
        super().__init__()
        self.self_attn = GptOssAttention(cfg, backend, layer_idx)
        self.mlp = GptOssMLP(cfg)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                                backend.rms_norm)

    def forward(self, x, cos, sin):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class GptOssForCausalLM(nn.Module):
    hf_architectures = ("GptOssForCausalLM",)
    config_class = GptOssConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> GptOssConfig:
        return GptOssConfig.from_hf_config(hf_cfg)

    def __init__(self, config: GptOssConfig | dict,
                 backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = GptOssConfig(**config)
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend),
            device_type, head_dim=config.head_dim,
        )
        self.config = config
        self.backend = backend
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(
            GptOssDecoderLayer(config, backend, i)
            for i in range(config.num_hidden_layers))
        inner.norm = RMSNorm(config.hidden_size, config.rms_norm_eps, backend.rms_norm)
        cos, sin = build_rope_cache(config.head_dim, config.max_position_embeddings,
                                    config.rope_theta, config.rope_scaling)
        inner.register_buffer("rope_cos", cos, persistent=False)
        inner.register_buffer("rope_sin", sin, persistent=False)
        self.model = inner
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids: torch.Tensor, labels: torch.Tensor | None = None,
                position_ids: torch.Tensor | None = None, **_: Any) -> torch.Tensor:
        m = self.model
        x = m.embed_tokens(input_ids)
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = m.rope_cos[:S], m.rope_sin[:S]
        else:
            cos, sin = m.rope_cos[position_ids[0]], m.rope_sin[position_ids[0]]
        if cos.dtype != torch.float32:
            cos, sin = cos.float(), sin.float()
        for layer in m.layers:
            x = layer(x, cos, sin)
        hidden = m.norm(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before passing labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            cos, sin = build_rope_cache(self.config.head_dim,
                                        self.config.max_position_embeddings,
                                        self.config.rope_theta,
                                        self.config.rope_scaling)
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
            elif isinstance(mod, GptOssExperts):
                for p in (mod.gate_up_proj, mod.down_proj):
                    nn.init.normal_(p, std=std)
                nn.init.zeros_(mod.gate_up_proj_bias)
                nn.init.zeros_(mod.down_proj_bias)
            elif isinstance(mod, GptOssAttention):
                nn.init.zeros_(mod.sinks)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
