"""MoE causal LM (Qwen3-MoE / Mixtral-class), MI355X-native.

Reference behavior: nemo_automodel/components/models/qwen3_moe/model.py and
components/moe/layers.py — llama-style attention + MoE FFN with stacked
expert weights, HF key parity via state_dict_adapter.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.llama.model import LlamaAttention, LlamaConfig
from automodel_amd.moe.config import MoEConfig
from automodel_amd.moe.layers import MoE, Gate, SharedExpert
from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import build_rope_cache


@dataclass
class MoEModelConfig(LlamaConfig):
    moe: MoEConfig = field(default_factory=MoEConfig)
    hf_flavor: str = "qwen3_moe"   # qwen3_moe | mixtral | qwen2_moe | ernie
    first_k_dense: int = 0         # leading dense (non-MoE) decoder layers

    def __post_init__(self):
        super().__post_init__()
        if isinstance(self.moe, dict):
            self.moe = MoEConfig(**self.moe)

    @classmethod
    def from_hf_config(cls, hf: Any) -> "MoEModelConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        base = LlamaConfig.from_hf_config(hf).__dict__.copy()
        arch = (hf.get("architectures") or ["Qwen3MoeForCausalLM"])[0]
        if "Mixtral" in arch:
            moe = MoEConfig(
                n_routed_experts=hf.get("num_local_experts", 8),
                n_activated_experts=hf.get("num_experts_per_tok", 2),
                moe_intermediate_size=hf.get("intermediate_size", 14336),
                aux_loss_coeff=hf.get("router_aux_loss_coef", 0.0),
                norm_topk_prob=True,
            )
            flavor = "mixtral"
        elif "Olmoe" in arch:
            moe = MoEConfig(
                n_routed_experts=hf.get("num_experts", 64),
                n_activated_experts=hf.get("num_experts_per_tok", 8),
                moe_intermediate_size=hf.get("intermediate_size", 1024),
                aux_loss_coeff=hf.get("router_aux_loss_coef", 0.0),
                norm_topk_prob=hf.get("norm_topk_prob", False),
            )
            base["qk_norm_full"] = True
            flavor = "qwen3_moe"   # same stacked-expert key layout
        elif "GraniteMoe" in arch:
            moe = MoEConfig(
                n_routed_experts=hf.get("num_local_experts", 8),
                n_activated_experts=hf.get("num_experts_per_tok", 2),
                moe_intermediate_size=hf.get("intermediate_size", 1024),
                aux_loss_coeff=hf.get("router_aux_loss_coef", 0.0),
                topk_then_softmax=True,
            )
            flavor = "qwen3_moe"   # stacked-key layout (+router rename)
        elif "MiniMaxM2" in arch:
            # sigmoid scores + aux-free correction bias (selection biased,
            # gathered weights unbiased, top-k renormalized) + FULL-width
            # q/k projection norms — rides the OLMoE attention variant
            moe = MoEConfig(
                n_routed_experts=hf.get("num_local_experts", 256),
                n_activated_experts=hf.get("num_experts_per_tok", 8),
                moe_intermediate_size=hf.get("intermediate_size", 1536),
                score_func="sigmoid",
                expert_bias=True,
                norm_topk_prob=True,
                aux_loss_coeff=hf.get("router_aux_loss_coef", 0.0),
            )
            base["qk_norm_full"] = True
            flavor = "qwen3_moe"
        elif "FlexOlmo" in arch:
            # OLMo-2 POST-norm layout + full-width q/k norms + softmax-topk
            moe = MoEConfig(
                n_routed_experts=hf.get("num_experts", 7),
                n_activated_experts=hf.get("num_experts_per_tok", 5),
                moe_intermediate_size=hf.get("intermediate_size", 11008),
                aux_loss_coeff=hf.get("router_aux_loss_coef", 0.0),
                norm_topk_prob=hf.get("norm_topk_prob", False),
            )
            base["qk_norm_full"] = True
            base["olmo2_layout"] = True
            return cls(**base, moe=moe, hf_flavor="qwen3_moe")
        elif "Dots1" in arch:
            # dots.llm1: DeepSeek-style sigmoid + aux-free-bias routing
            # (group top-k, shared expert, dense-first layers) over a
            # qwen3-style per-head qk-norm attention
            n_shared = hf.get("n_shared_experts") or 0
            moe = MoEConfig(
                n_routed_experts=hf.get("n_routed_experts") or 64,
                n_activated_experts=hf.get("num_experts_per_tok") or 8,
                moe_intermediate_size=hf.get("moe_intermediate_size", 1408),
                n_shared_experts=n_shared,
                shared_expert_intermediate_size=(
                    hf.get("moe_intermediate_size", 1408) * n_shared or None),
                score_func="sigmoid",
                expert_bias=True,
                norm_topk_prob=hf.get("norm_topk_prob", False),
                route_scale=hf.get("routed_scaling_factor", 1.0),
                n_expert_groups=hf.get("n_group", 1) or 1,
                n_limited_groups=hf.get("topk_group", 1) or 1,
            )
            base["qk_norm"] = True
            return cls(**base, moe=moe, hf_flavor="qwen3_moe",
                       first_k_dense=hf.get("first_k_dense_replace", 0))
        elif "Qwen2Moe" in arch:
            moe = MoEConfig(
                n_routed_experts=hf.get("num_experts", 60),
                n_activated_experts=hf.get("num_experts_per_tok", 4),
                n_shared_experts=1,
                shared_expert_intermediate_size=hf.get(
                    "shared_expert_intermediate_size", 5632),
                shared_expert_gate=True,
                moe_intermediate_size=hf.get("moe_intermediate_size", 1408),
                aux_loss_coeff=hf.get("router_aux_loss_coef", 0.0),
                norm_topk_prob=hf.get("norm_topk_prob", False),
            )
            flavor = "qwen2_moe"
        elif "Ernie4_5_Moe" in arch:
            # softmax scores; selection biased by moe_statics correction
            # bias, weights gathered unbiased then renormalized
            n_shared = hf.get("moe_num_shared_experts", 0)
            moe = MoEConfig(
                n_routed_experts=hf.get("moe_num_experts", 64),
                n_activated_experts=hf.get("moe_k", 8),
                moe_intermediate_size=hf.get("moe_intermediate_size", 768),
                n_shared_experts=1 if n_shared else 0,
                shared_expert_intermediate_size=(
                    hf.get("moe_intermediate_size", 768) * n_shared or None),
                expert_bias=True,
                norm_topk_prob=True,
                aux_loss_coeff=hf.get("router_aux_loss_coef", 0.0),
            )
            flavor = "ernie"
            return cls(**base, moe=moe, hf_flavor=flavor,
                       first_k_dense=hf.get("moe_layer_start_index", 0))
        else:
            moe = MoEConfig(
                n_routed_experts=hf.get("num_experts", 64),
                n_activated_experts=hf.get("num_experts_per_tok", 8),
                moe_intermediate_size=hf.get("moe_intermediate_size", 768),
                aux_loss_coeff=hf.get("router_aux_loss_coef", 0.0),
                norm_topk_prob=hf.get("norm_topk_prob", True),
            )
            flavor = "qwen3_moe"
        return cls(**base, moe=moe, hf_flavor=flavor)


class MoEDecoderLayer(nn.Module):
    def __init__(self, cfg: MoEModelConfig, backend: BackendConfig, dense: bool = False):
        super().__init__()
        self.cfg = cfg
        self.self_attn = LlamaAttention(cfg, backend)
        # leading dense layers (ernie/deepseek first_k_dense): plain SwiGLU
        self.mlp = (SharedExpert(cfg.hidden_size, cfg.intermediate_size)
                    if dense else MoE(cfg.hidden_size, cfg.moe))
        if cfg.olmo2_layout:   # FlexOlmo: POST-norm after each sublayer
            self.post_attention_layernorm = RMSNorm(
                cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
            self.post_feedforward_layernorm = RMSNorm(
                cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        else:
            self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                           backend.rms_norm)
            self.post_attention_layernorm = RMSNorm(
                cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)

    def forward(self, x, cos, sin):
        r = self.cfg.residual_multiplier
        if self.cfg.olmo2_layout:
            x = x + self.post_attention_layernorm(self.self_attn(x, cos, sin)) * r
            return x + self.post_feedforward_layernorm(self.mlp(x)) * r
        x = x + self.self_attn(self.input_layernorm(x), cos, sin) * r
        x = x + self.mlp(self.post_attention_layernorm(x)) * r
        return x


class MoEForCausalLM(nn.Module):
    hf_architectures = ("Qwen3MoeForCausalLM", "Qwen2MoeForCausalLM", "MixtralForCausalLM",
                        "OlmoeForCausalLM", "GraniteMoeForCausalLM",
                        "Ernie4_5_MoeForCausalLM", "MiniMaxM2ForCausalLM",
                        "Dots1ForCausalLM", "FlexOlmoForCausalLM")
    config_class = MoEModelConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> MoEModelConfig:
        return MoEModelConfig.from_hf_config(hf_cfg)

    def __init__(self, config: MoEModelConfig | dict, backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = MoEModelConfig(**config)
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend), device_type
        )
        self.config = config
        self.backend = backend
        self.model = nn.Module()
        self.model.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.model.layers = nn.ModuleList(
            MoEDecoderLayer(config, backend, dense=i < config.first_k_dense)
            for i in range(config.num_hidden_layers)
        )
        self.model.norm = RMSNorm(config.hidden_size, config.rms_norm_eps, backend.rms_norm)
        cos, sin = build_rope_cache(config.head_dim, config.max_position_embeddings,
                                    config.rope_theta, config.rope_scaling)
        self.model.register_buffer("rope_cos", cos, persistent=False)
        self.model.register_buffer("rope_sin", sin, persistent=False)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.loss_fn = None
        self.state_dict_adapter = MoEStateDictAdapter(config)

    def forward(self, input_ids, labels=None, position_ids=None, return_hidden=False, **_):
        x = self.model.embed_tokens(input_ids)
        if self.config.embedding_multiplier != 1.0:    # granite-moe
            x = x * self.config.embedding_multiplier
        S = input_ids.shape[1]
        cos, sin = self.model.rope_cos[:S], self.model.rope_sin[:S]
        if cos.dtype != torch.float32:
            cos, sin = cos.float(), sin.float()
        for layer in self.model.layers:
            x = layer(x, cos, sin)
        x = self.model.norm(x)
        if self.config.logits_scaling != 1.0 and not return_hidden:
            x = x / self.config.logits_scaling
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
            aux = getattr(gate, "last_aux_loss", None)
            if aux is not None:
                total = aux if total is None else total + aux
                layer.mlp.gate.last_aux_loss = None
        return total

    @torch.no_grad()
    def update_moe_gate_bias(self) -> None:
        """Aux-free balancing bias update after each optim step
        (reference train_ft.py update_moe_gate_bias)."""
        for layer in self.model.layers:
            load = getattr(layer.mlp, "last_expert_load", None)
            if load is not None:
                layer.mlp.gate.update_bias(load)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        std = self.config.initializer_range
        if device is not None:
            self.to_empty(device=device)
            cos, sin = build_rope_cache(self.config.head_dim,
                                        self.config.max_position_embeddings,
                                        self.config.rope_theta,
                                        self.config.rope_scaling, device=device)
            self.model.rope_cos.copy_(cos)
            self.model.rope_sin.copy_(sin)
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, RMSNorm):
                nn.init.ones_(m.weight)
            elif isinstance(m, Gate):
                nn.init.normal_(m.weight, std=std)
        for m in self.modules():
            if type(m).__name__ == "GroupedExperts":
                m.init_weights(std)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        seen, total = set(), 0
        for p in self.parameters():
            if id(p) not in seen:
                seen.add(id(p))
                total += p.numel()
        return total
