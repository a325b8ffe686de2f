"""Gemma-family causal LM (Gemma-2-style architecture), MI355X-native.

Reference behavior: the reference ships gemma model families
(nemo_automodel/components/models/gemma4_*/model.py wrap HF gemma classes);
here the architecture is implemented directly. Gemma-2 deltas vs Llama:

  * GeGLU MLP (tanh-gelu gate) instead of SwiGLU;
  * RMSNorm computes x_hat * (1 + w) with zero-init w;
  * embeddings scaled by sqrt(hidden_size);
  * four norms per layer (pre/post attention, pre/post MLP);
  * optional attn/final logit soft-capping (tanh);
  * sliding-window attention on alternating layers;
  * decoupled head_dim (256 for gemma-2-9b) — attention rides sdpa via the
    BackendConfig head-dim resolve (ops flash kernel is tiled for 128).

HIP kernel reuse: rope runs the in-tree kernel; norms reuse the HIP rms_norm
with the (1+w) scale folded at call time; GeGLU is elementwise (bandwidth-
bound, fused by eager into two kernels).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rms_norm import rms_norm
from automodel_amd.ops.rope import apply_rope, build_rope_cache


@dataclass
class GemmaConfig:
    vocab_size: int = 256000
    hidden_size: int = 2304
    intermediate_size: int = 9216
    num_hidden_layers: int = 26
    num_attention_heads: int = 8
    num_key_value_heads: int = 4
    head_dim: int = 256
    max_position_embeddings: int = 8192
    rope_theta: float = 10000.0
    rms_norm_eps: float = 1e-6
    attn_logit_softcapping: float | None = 50.0
    final_logit_softcapping: float | None = 30.0
    sliding_window: int | None = 4096
    query_pre_attn_scalar: float | None = 256.0
    tie_word_embeddings: bool = True
    # --- gemma-3 extensions (models/gemma/ covers both generations) ---
    qk_norm: bool = False                 # per-head (1+w) q/k RMSNorm
    layer_types: list | None = None       # explicit sliding/full pattern
    rope_local_base_freq: float | None = None  # sliding layers' theta
    rope_scaling: dict | None = None      # global layers' scaling (linear 8x)
    post_norms: bool = True               # gemma-2/3: extra post-attn/ffn norms
                                          # (gemma-1 has only the two pre-norms)

    @classmethod
    def from_hf_config(cls, hf: dict) -> "GemmaConfig":
        get = hf.get
        archs = " ".join(get("architectures", []) or [])
        v1 = bool(archs) and "Gemma2" not in archs and "Gemma3" not in archs
        return cls(
            post_norms=not v1,
            vocab_size=get("vocab_size", 256000),
            hidden_size=get("hidden_size", 2304),
            intermediate_size=get("intermediate_size", 9216),
            num_hidden_layers=get("num_hidden_layers", 26),
            num_attention_heads=get("num_attention_heads", 8),
            num_key_value_heads=get("num_key_value_heads", 4),
            head_dim=get("head_dim", 256),
            max_position_embeddings=get("max_position_embeddings", 8192),
            rope_theta=get("rope_theta", 10000.0),
            rms_norm_eps=get("rms_norm_eps", 1e-6),
            attn_logit_softcapping=None if v1 else get("attn_logit_softcapping", 50.0),
            final_logit_softcapping=None if v1 else get("final_logit_softcapping", 30.0),
            sliding_window=None if v1 else get("sliding_window", 4096),
            query_pre_attn_scalar=None if v1 else get("query_pre_attn_scalar", 256.0),
            tie_word_embeddings=get("tie_word_embeddings", True),
        )

    @classmethod
    def from_hf_gemma3(cls, hf: dict) -> "GemmaConfig":
        """Gemma3Text configs: per-layer-type rope_parameters, qk-norm, no
        softcapping."""
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or g("rope_scaling") or {}
        local = (rp.get("sliding_attention") or {})
        full = (rp.get("full_attention") or {})
        cfg = cls.from_hf_config(hf)
        cfg.qk_norm = True
        cfg.attn_logit_softcapping = None
        cfg.final_logit_softcapping = None
        cfg.layer_types = g("layer_types")
        cfg.rope_local_base_freq = local.get("rope_theta",
                                             g("rope_local_base_freq", 10000.0))
        cfg.rope_theta = full.get("rope_theta", g("rope_theta", 1000000.0))
        if full.get("rope_type") == "linear":
            cfg.rope_scaling = {"rope_type": "linear", "factor": full.get("factor", 8.0)}
        return cfg


class GemmaRMSNorm(nn.Module):
    """x_hat * (1 + w), zero-init w (HF Gemma convention). Folds into the
    HIP rms_norm kernel by passing (1 + w) as the scale."""

    def __init__(self, dim: int, eps: float, backend: str):
        super().__init__()
        self.weight = nn.Parameter(torch.zeros(dim))
        self.eps = eps
        self.backend = backend

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return rms_norm(x, 1.0 + self.weight, self.eps, backend=self.backend)


def _softcap(x: torch.Tensor, cap: float | None) -> torch.Tensor:
    return x if cap is None else torch.tanh(x / cap) * cap


class GemmaAttention(nn.Module):
    def __init__(self, cfg: GemmaConfig, backend: BackendConfig, layer_idx: int):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.num_heads, self.num_kv_heads, self.head_dim = H, Hk, D
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=False)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        self.scale = (cfg.query_pre_attn_scalar or D) ** -0.5
        self.softcap = cfg.attn_logit_softcapping
        if cfg.layer_types is not None:          # gemma-3 explicit pattern
            self.is_sliding = cfg.layer_types[layer_idx] == "sliding_attention"
        else:                                    # gemma-2: even layers sliding
            self.is_sliding = layer_idx % 2 == 0
        self.window = cfg.sliding_window if self.is_sliding else None
        if cfg.qk_norm:
            self.q_norm = GemmaRMSNorm(D, cfg.rms_norm_eps, backend.rms_norm)
            self.k_norm = GemmaRMSNorm(D, cfg.rms_norm_eps, backend.rms_norm)
        self.qk_norm = cfg.qk_norm
        self.backend = backend

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                block_ids: torch.Tensor | None = None) -> torch.Tensor:
        B, S, _ = x.shape
        q = self.q_proj(x).view(B, S, -1, self.head_dim)
        k = self.k_proj(x).view(B, S, -1, self.head_dim)
        v = self.v_proj(x).view(B, S, -1, self.head_dim)
        if self.qk_norm:
            q = self.q_norm(q)
            k = self.k_norm(k)
        q, k = apply_rope(q, k, cos, sin, backend=self.backend.rope)
        if self.softcap is None and self.window is None and block_ids is None:
            o = flash_attention(q, k, v, causal=True, scale=self.scale,
                                backend=self.backend.attn)
        else:
            o = self._eager_capped(q, k, v, block_ids)
        return self.o_proj(o.reshape(B, S, -1))

    def _eager_capped(self, q, k, v, block_ids=None):
        """Softcap/sliding-window need the scores; bandwidth-bound eager path
        (these models route here only for short eval shapes; long-context
        gemma training uses the global layers' flash path)."""
        B, S, H, D = q.shape
        qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))
        rep = H // self.num_kv_heads
        if rep > 1:
            kt = kt.repeat_interleave(rep, dim=1)
            vt = vt.repeat_interleave(rep, dim=1)
        scores = torch.einsum("bhqd,bhkd->bhqk", qt.float(), kt.float()) * self.scale
        scores = _softcap(scores, self.softcap)
        i = torch.arange(S, device=q.device)
        allowed = i[None, :] <= i[:, None]                # causal
        if block_ids is not None:
            # gemma-3 multimodal: image tokens attend bidirectionally
            # within their own image block
            same = (block_ids[:, :, None] == block_ids[:, None, :]) \
                & (block_ids[:, :, None] >= 0)
            allowed = allowed[None] | same
        else:
            allowed = allowed[None].expand(1, S, S)
        if self.window is not None:
            within = (i[None, :] - i[:, None]).abs() < self.window
            allowed = allowed & within[None]
        scores = scores.masked_fill(~allowed.unsqueeze(1), float("-inf"))
        o = torch.einsum("bhqk,bhkd->bhqd", scores.softmax(-1), vt.float())
        return o.transpose(1, 2).to(q.dtype)


class GemmaMLP(nn.Module):
    def __init__(self, cfg: GemmaConfig):
        super().__init__()
        self.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        self.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.down_proj(F.gelu(self.gate_proj(x), approximate="tanh") * self.up_proj(x))


class GemmaDecoderLayer(nn.Module):
    def __init__(self, cfg: GemmaConfig, backend: BackendConfig, layer_idx: int):
        super().__init__()
        nb = backend.rms_norm
        self.post_norms = cfg.post_norms
        self.self_attn = GemmaAttention(cfg, backend, layer_idx)
        self.mlp = GemmaMLP(cfg)
        self.input_layernorm = GemmaRMSNorm(cfg.hidden_size, cfg.rms_norm_eps, nb)
        self.post_attention_layernorm = GemmaRMSNorm(cfg.hidden_size, cfg.rms_norm_eps, nb)
        if cfg.post_norms:
            self.pre_feedforward_layernorm = GemmaRMSNorm(cfg.hidden_size, cfg.rms_norm_eps, nb)
            self.post_feedforward_layernorm = GemmaRMSNorm(cfg.hidden_size, cfg.rms_norm_eps, nb)

    def forward(self, x, cos, sin, cos_local=None, sin_local=None,
                block_ids=None):
        if cos_local is not None and self.self_attn.is_sliding:
            cos, sin = cos_local, sin_local   # gemma-3 dual-frequency rope
        if not self.post_norms:   # gemma-1: classic pre-norm residual order
            x = x + self.self_attn(self.input_layernorm(x), cos, sin, block_ids)
            return x + self.mlp(self.post_attention_layernorm(x))
        x = x + self.post_attention_layernorm(
            self.self_attn(self.input_layernorm(x), cos, sin, block_ids))
        x = x + self.post_feedforward_layernorm(self.mlp(self.pre_feedforward_layernorm(x)))
        return x


class GemmaForCausalLM(nn.Module):
    hf_architectures = ("Gemma2ForCausalLM", "GemmaForCausalLM")
    config_class = GemmaConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> GemmaConfig:
        return GemmaConfig.from_hf_config(hf_cfg)

    def __init__(self, config: GemmaConfig | dict, backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = GemmaConfig(**config)
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend),
            device_type, head_dim=config.head_dim,
        )
        self.config = config
        self.backend = backend
        self.model = nn.ModuleDict()  # placeholder replaced below for HF key parity
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(
            GemmaDecoderLayer(config, backend, i) for i in range(config.num_hidden_layers)
        )
        inner.norm = GemmaRMSNorm(config.hidden_size, config.rms_norm_eps, backend.rms_norm)
        cos, sin = build_rope_cache(config.head_dim, config.max_position_embeddings,
                                    config.rope_theta, config.rope_scaling)
        inner.register_buffer("rope_cos", cos, persistent=False)
        inner.register_buffer("rope_sin", sin, persistent=False)
        if config.rope_local_base_freq is not None:
            cl, sl = build_rope_cache(config.head_dim,
                                      config.max_position_embeddings,
                                      config.rope_local_base_freq)
            inner.register_buffer("rope_cos_local", cl, persistent=False)
            inner.register_buffer("rope_sin_local", sl, persistent=False)
        else:
            inner.rope_cos_local = None
            inner.rope_sin_local = None
        self.model = inner
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids: torch.Tensor, labels: torch.Tensor | None = None,
                position_ids: torch.Tensor | None = None,
                inputs_embeds: torch.Tensor | None = None,
                block_ids: torch.Tensor | None = None, **_: Any) -> torch.Tensor:
        m = self.model
        base = inputs_embeds if inputs_embeds is not None else m.embed_tokens(input_ids)
        x = base * (self.config.hidden_size ** 0.5)
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = m.rope_cos[:S], m.rope_sin[:S]
        else:
            cos, sin = m.rope_cos[position_ids[0]], m.rope_sin[position_ids[0]]
        if cos.dtype != torch.float32:
            cos, sin = cos.float(), sin.float()
        cos_l = sin_l = None
        if m.rope_cos_local is not None:
            if position_ids is None:
                cos_l, sin_l = m.rope_cos_local[:S], m.rope_sin_local[:S]
            else:
                cos_l = m.rope_cos_local[position_ids[0]]
                sin_l = m.rope_sin_local[position_ids[0]]
            if cos_l.dtype != torch.float32:
                cos_l, sin_l = cos_l.float(), sin_l.float()
        for layer in m.layers:
            x = layer(x, cos, sin, cos_l, sin_l, block_ids)
        hidden = m.norm(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before passing labels"
            if self.config.final_logit_softcapping is not None:
                logits = _softcap(self.lm_head(hidden), self.config.final_logit_softcapping)
                V = logits.shape[-1]
                return F.cross_entropy(logits.reshape(-1, V).float(),
                                       labels.reshape(-1), ignore_index=-100,
                                       reduction="sum")
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return _softcap(self.lm_head(hidden), self.config.final_logit_softcapping)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        std = 0.02
        if device is not None:
            self.to_empty(device=device)
            m = self.model
            cos, sin = build_rope_cache(self.config.head_dim,
                                        self.config.max_position_embeddings,
                                        self.config.rope_theta)
            m.rope_cos.copy_(cos.to(m.rope_cos.device))
            m.rope_sin.copy_(sin.to(m.rope_sin.device))
        for mod in self.modules():
            if isinstance(mod, nn.Linear):
                nn.init.normal_(mod.weight, mean=0.0, std=std)
            elif isinstance(mod, nn.Embedding):
                nn.init.normal_(mod.weight, mean=0.0, std=std)
            elif isinstance(mod, GemmaRMSNorm):
                nn.init.zeros_(mod.weight)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight


class Gemma3ForCausalLM(GemmaForCausalLM):
    """Gemma-3 text model: Gemma-2 base + qk-norm, explicit layer_types,
    dual-frequency rope (local theta on sliding layers), no softcapping."""

    hf_architectures = ("Gemma3ForCausalLM",)  # the ConditionalGeneration VLM variant is round-2

    @staticmethod
    def config_from_hf(hf_cfg) -> GemmaConfig:
        return GemmaConfig.from_hf_gemma3(hf_cfg)
