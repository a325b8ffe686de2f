"""DeepSeek-V3-style causal LM: MLA attention + MoE FFN (+dense first layers).

Reference behavior: nemo_automodel/components/models/deepseek_v3/
(model.py, layers.py:41 MLA, rope_utils.py YaRN). Multi-head Latent
Attention: queries and KV are projected through low-rank bottlenecks; RoPE is
applied to a decoupled rope sub-dimension (shared single k_rope head).

Attention head dims differ from 128 (qk = nope+rope, v = v_head_dim): the
attention core runs the split-dim (Dqk=192, Dv=128) instantiation of the
in-tree HIP flash kernel (round 2), with torch-SDPA as the CPU/odd-shape
fallback. All other hot ops (RMSNorm, SwiGLU, grouped experts, fused CE)
also run the in-tree HIP kernels.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.moe.config import MoEConfig
from automodel_amd.moe.layers import MoE
from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope, build_rope_cache
from automodel_amd.ops.swiglu import swiglu


@dataclass
class DeepseekV3Config:
    vocab_size: int = 129280
    hidden_size: int = 7168
    intermediate_size: int = 18432           # dense layers
    num_hidden_layers: int = 4
    num_attention_heads: int = 16
    first_k_dense_replace: int = 1           # dense layers before MoE starts
    q_lora_rank: int | None = 1536
    kv_lora_rank: int = 512
    qk_nope_head_dim: int = 128
    qk_rope_head_dim: int = 64
    v_head_dim: int = 128
    rms_norm_eps: float = 1e-6
    rope_theta: float = 10000.0
    rope_scaling: dict | None = None
    max_position_embeddings: int = 4096
    initializer_range: float = 0.02
    tie_word_embeddings: bool = False
    moe: MoEConfig = field(default_factory=lambda: MoEConfig(
        n_routed_experts=8, n_shared_experts=1, n_activated_experts=2,
        score_func="sigmoid", expert_bias=True, moe_intermediate_size=2048,
    ))

    def __post_init__(self):
        if isinstance(self.moe, dict):
            self.moe = MoEConfig(**self.moe)

    @property
    def qk_head_dim(self) -> int:
        return self.qk_nope_head_dim + self.qk_rope_head_dim

    @classmethod
    def from_hf_config(cls, hf: Any) -> "DeepseekV3Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        archs = " ".join(g("architectures", []) or [])
        # V2: softmax scoring, no aux-free bias; V3: sigmoid + bias
        v2 = "DeepseekV2" in archs
        return cls(
            vocab_size=g("vocab_size", 129280),
            hidden_size=g("hidden_size", 7168),
            intermediate_size=g("intermediate_size", 18432),
            num_hidden_layers=g("num_hidden_layers", 61),
            num_attention_heads=g("num_attention_heads", 128),
            first_k_dense_replace=g("first_k_dense_replace", 3),
            q_lora_rank=g("q_lora_rank", 1536),
            kv_lora_rank=g("kv_lora_rank", 512),
            qk_nope_head_dim=g("qk_nope_head_dim", 128),
            qk_rope_head_dim=g("qk_rope_head_dim", 64),
            v_head_dim=g("v_head_dim", 128),
            rms_norm_eps=g("rms_norm_eps", 1e-6),
            rope_theta=g("rope_theta", 10000.0),
            rope_scaling=g("rope_scaling"),
            max_position_embeddings=g("max_position_embeddings", 4096),
            moe=MoEConfig(
                n_routed_experts=g("n_routed_experts", 256),
                n_shared_experts=g("n_shared_experts", 1),
                n_activated_experts=g("num_experts_per_tok", 8),
                n_expert_groups=g("n_group", 1),
                n_limited_groups=g("topk_group", 1),
                score_func=g("scoring_func", "softmax" if v2 else "sigmoid"),
                # HF V2 router never renormalizes the top-k weights
                norm_topk_prob=False if v2 else g("norm_topk_prob", True),
                route_scale=g("routed_scaling_factor", 1.0),
                expert_bias=not v2,
                moe_intermediate_size=g("moe_intermediate_size", 2048),
                shared_expert_intermediate_size=(
                    g("moe_intermediate_size", 2048) * g("n_shared_experts", 1)
                ),
            ),
        )


class MLAAttention(nn.Module):
    """Multi-head Latent Attention (reference deepseek_v3/layers.py:41)."""

    def __init__(self, cfg: DeepseekV3Config, backend: BackendConfig):
        super().__init__()
        self.cfg = cfg
        self.backend = backend
        H = cfg.num_attention_heads
        self.num_heads = H
        if cfg.q_lora_rank:
            self.q_a_proj = nn.Linear(cfg.hidden_size, cfg.q_lora_rank, bias=False)
            self.q_a_layernorm = RMSNorm(cfg.q_lora_rank, cfg.rms_norm_eps, backend.rms_norm)
            self.q_b_proj = nn.Linear(cfg.q_lora_rank, H * cfg.qk_head_dim, bias=False)
        else:
            self.q_proj = nn.Linear(cfg.hidden_size, H * cfg.qk_head_dim, bias=False)
        self.kv_a_proj_with_mqa = nn.Linear(
            cfg.hidden_size, cfg.kv_lora_rank + cfg.qk_rope_head_dim, bias=False)
        self.kv_a_layernorm = RMSNorm(cfg.kv_lora_rank, cfg.rms_norm_eps, backend.rms_norm)
        self.kv_b_proj = nn.Linear(
            cfg.kv_lora_rank, H * (cfg.qk_nope_head_dim + cfg.v_head_dim), bias=False)
        self.o_proj = nn.Linear(H * cfg.v_head_dim, cfg.hidden_size, bias=False)
        self.scale = cfg.qk_head_dim**-0.5

    def _qkv(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor):
        """Project to the decoupled-rope (qf, kf, v) triple [B,S,H,D*]."""
        cfg = self.cfg
        B, S, _ = x.shape
        H = self.num_heads
        if cfg.q_lora_rank:
            q = self.q_b_proj(self.q_a_layernorm(self.q_a_proj(x)))
        else:
            q = self.q_proj(x)
        q = q.view(B, S, H, cfg.qk_head_dim)
        q_nope, q_rope = q.split([cfg.qk_nope_head_dim, cfg.qk_rope_head_dim], dim=-1)

        kv_a = self.kv_a_proj_with_mqa(x)
        kv_c, k_rope = kv_a.split([cfg.kv_lora_rank, cfg.qk_rope_head_dim], dim=-1)
        kv = self.kv_b_proj(self.kv_a_layernorm(kv_c)).view(
            B, S, H, cfg.qk_nope_head_dim + cfg.v_head_dim)
        k_nope, v = kv.split([cfg.qk_nope_head_dim, cfg.v_head_dim], dim=-1)

        # decoupled rope: q_rope per head, k_rope single shared head.
        # DeepSeek checkpoints store the rope dims INTERLEAVED ([x0,y0,x1,y1..]);
        # de-interleave to half-split order before the rotate-half kernel
        # (HF apply_rotary_pos_emb_interleave equivalent).
        k_rope = k_rope.view(B, S, 1, cfg.qk_rope_head_dim)
        d2 = cfg.qk_rope_head_dim // 2
        q_rope = q_rope.view(B, S, H, d2, 2).transpose(-1, -2).reshape(B, S, H, -1)
        k_rope = k_rope.view(B, S, 1, d2, 2).transpose(-1, -2).reshape(B, S, 1, -1)
        q_rope, k_rope = apply_rope(q_rope.contiguous(), k_rope.contiguous(),
                                    cos, sin, backend="torch")
        k_rope = k_rope.expand(B, S, H, cfg.qk_rope_head_dim)

        qf = torch.cat([q_nope, q_rope], dim=-1)   # B,S,H,qk (192 = nope+rope)
        kf = torch.cat([k_nope, k_rope], dim=-1)
        return qf, kf, v

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        cfg = self.cfg
        B, S, _ = x.shape
        H = self.num_heads
        qf, kf, v = self._qkv(x, cos, sin)
        if self.backend.attn == "hip" and qf.is_cuda:
            # split-dim flash kernel: (Dqk=192, Dv=128) instantiation
            from automodel_amd.ops.attention import flash_attention

            o = flash_attention(qf, kf, v.contiguous(), causal=True,
                                scale=self.scale, backend="hip")
            return self.o_proj(o.reshape(B, S, H * cfg.v_head_dim))
        o = torch.nn.functional.scaled_dot_product_attention(
            qf.transpose(1, 2), kf.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, scale=self.scale)
        return self.o_proj(o.transpose(1, 2).reshape(B, S, H * cfg.v_head_dim))


class DenseMLP(nn.Module):
    def __init__(self, hidden: int, inter: int):
        super().__init__()
        self.gate_proj = nn.Linear(hidden, inter, bias=False)
        self.up_proj = nn.Linear(hidden, inter, bias=False)
        self.down_proj = nn.Linear(inter, hidden, bias=False)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


class DeepseekV3DecoderLayer(nn.Module):
    def __init__(self, cfg: DeepseekV3Config, backend: BackendConfig, layer_idx: int):
        super().__init__()
        self.self_attn = MLAAttention(cfg, backend)
        if layer_idx < cfg.first_k_dense_replace:
            self.mlp = DenseMLP(cfg.hidden_size, cfg.intermediate_size)
        else:
            self.mlp = MoE(cfg.hidden_size, cfg.moe)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                                backend.rms_norm)

    def forward(self, x, cos, sin):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class DeepseekV3ForCausalLM(nn.Module):
    # Kimi-K2 is DeepSeek-V3-architecture-compatible (reference
    # models/kimi_k2/config.py: KimiK2Config(DeepseekV3Config)); Moonlight
    # ships the same architecture string
    hf_architectures = ("DeepseekV3ForCausalLM", "DeepseekV2ForCausalLM",
                        "KimiK2ForCausalLM")
    config_class = DeepseekV3Config

    @staticmethod
    def config_from_hf(hf_cfg) -> DeepseekV3Config:
        return DeepseekV3Config.from_hf_config(hf_cfg)

    def __init__(self, config: DeepseekV3Config | dict,
                 backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = DeepseekV3Config(**config)
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend),
            device_type)
        self.config = config
        self.backend = backend
        self.model = nn.Module()
        self.model.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.model.layers = nn.ModuleList(
            DeepseekV3DecoderLayer(config, backend, i)
            for i in range(config.num_hidden_layers))
        self.model.norm = RMSNorm(config.hidden_size, config.rms_norm_eps, backend.rms_norm)
        cos, sin = build_rope_cache(config.qk_rope_head_dim,
                                    config.max_position_embeddings,
                                    config.rope_theta, config.rope_scaling)
        self.model.register_buffer("rope_cos", cos, persistent=False)
        self.model.register_buffer("rope_sin", sin, persistent=False)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        self.loss_fn = None
        self.state_dict_adapter = MoEStateDictAdapter(config)

    def forward(self, input_ids, labels=None, position_ids=None, return_hidden=False, **_):
        x = self.model.embed_tokens(input_ids)
        S = input_ids.shape[1]
        cos, sin = self.model.rope_cos[:S].float(), self.model.rope_sin[:S].float()
        for layer in self.model.layers:
            x = layer(x, cos, sin)
        x = self.model.norm(x)
        if labels is not None:
            assert self.loss_fn is not None
            loss = self.loss_fn(x, self.lm_head.weight, labels)
            aux = self.collect_aux_losses()
            return loss + aux if aux is not None else loss
        if return_hidden:
            return x
        return self.lm_head(x)

    def collect_aux_losses(self):
        total = None
        for layer in self.model.layers:
            gate = getattr(layer.mlp, "gate", None)
            aux = getattr(gate, "last_aux_loss", None) if gate is not None else None
            if aux is not None:
                total = aux if total is None else total + aux
                layer.mlp.gate.last_aux_loss = None
        return total

    @torch.no_grad()
    def update_moe_gate_bias(self) -> None:
        for layer in self.model.layers:
            if isinstance(layer.mlp, MoE) and layer.mlp.last_expert_load is not None:
                layer.mlp.gate.update_bias(layer.mlp.last_expert_load)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        from automodel_amd.moe.layers import Gate

        std = self.config.initializer_range
        if device is not None:
            self.to_empty(device=device)
            cos, sin = build_rope_cache(self.config.qk_rope_head_dim,
                                        self.config.max_position_embeddings,
                                        self.config.rope_theta,
                                        self.config.rope_scaling, device=device)
            self.model.rope_cos.copy_(cos)
            self.model.rope_sin.copy_(sin)
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, RMSNorm):
                nn.init.ones_(m.weight)
            elif isinstance(m, Gate):
                nn.init.normal_(m.weight, std=std)
                if m.cfg.expert_bias:
                    m.e_score_correction_bias.zero_()
        for m in self.modules():
            if type(m).__name__ == "GroupedExperts":
                m.init_weights(std)

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
