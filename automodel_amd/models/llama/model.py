"""Llama-family causal LM, MI355X-native.

Custom DTensor-friendly implementation in the spirit of the reference's
in-tree model (reference: nemo_automodel/components/models/llama/model.py),
with kernels selected by BackendConfig: HIP flash attention, HIP RMSNorm,
fused HIP RoPE, and (via the recipe's loss) fused linear cross-entropy.
Parameter names match the HF checkpoint layout exactly
(model.layers.N.self_attn.q_proj.weight, ...) so the state-dict adapter is
the identity for this family.

Also covers Qwen2-style variants (attention bias) via LlamaConfig flags.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope, build_rope_cache
from automodel_amd.ops.swiglu import swiglu, swiglu_cat


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    head_dim: int | None = None
    max_position_embeddings: int = 8192
    rms_norm_eps: float = 1e-5
    rope_theta: float = 500000.0
    rope_scaling: dict | None = None
    tie_word_embeddings: bool = False
    attention_bias: bool = False
    mlp_bias: bool = False
    qk_norm: bool = False          # Qwen3-style per-head q/k RMSNorm
    qk_norm_after_rope: bool = False   # HunYuan: per-head norm AFTER rope
    qk_norm_full: bool = False     # OLMoE-style full-projection q/k RMSNorm
    mlp_type: str = "swiglu"       # "swiglu" | "xielu" (Apertus: up->xIELU->down)
    olmo2_layout: bool = False     # OLMo-2: norms on sublayer OUTPUTS only
    # granite-style scalar multipliers (neutral defaults)
    attention_multiplier: float | None = None   # attention scale override
    residual_multiplier: float = 1.0
    embedding_multiplier: float = 1.0
    logits_scaling: float = 1.0
    bidirectional: bool = False    # no causal mask (retrieval embedding models,
                                   # reference models/llama_bidirectional/)
    no_rope_layers: list | None = None  # SmolLM3 NoPE: per-layer 1=rope, 0=skip
    sliding_window: int | None = None   # Mistral-style windowed causal attention
    rope_interleaved: bool = False      # GPT-NeoX/Ernie pair-interleaved rope
    fused_qkv: bool = False        # one qkv GEMM (state_dict_adapter keeps HF keys)
    fused_gate_up: bool = False    # one gate|up GEMM + concatenated swiglu
    # HF-key substring renames (hunyuan/apertus name their norms differently)
    hf_key_renames: dict | None = None
    initializer_range: float = 0.02

    def __post_init__(self):
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "LlamaConfig":
        """Build from an HF config object or dict (llama / qwen2 / mistral)."""
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        get = hf.get
        archs = " ".join(get("architectures", []) or [])
        # newer transformers store theta/scaling under rope_parameters
        rp = get("rope_parameters") or {}
        theta = rp.get("rope_theta", get("rope_theta", 10000.0))
        scaling = get("rope_scaling")
        if scaling is None and rp.get("rope_type", "default") != "default":
            scaling = rp
        return cls(
            qk_norm=("Qwen3" in archs) or ("Apertus" in archs)
                or ("HunYuanDense" in archs),
            qk_norm_after_rope="HunYuanDense" in archs,
            mlp_type="xielu" if "Apertus" in archs else "swiglu",
            hf_key_renames=(
                {"query_layernorm": "q_norm", "key_layernorm": "k_norm"}
                if "HunYuanDense" in archs else
                {"attention_layernorm": "input_layernorm",
                 "feedforward_layernorm": "post_attention_layernorm"}
                if "Apertus" in archs else None),
            qk_norm_full="Olmo" in archs,
            olmo2_layout="Olmo2" in archs,
            # Phi-3 ships fused qkv_proj / gate_up_proj weights — exactly this
            # family's fused layout (keys and math match; parity-tested)
            fused_qkv="Phi3" in archs,
            fused_gate_up="Phi3" in archs,
            vocab_size=get("vocab_size", 32000),
            hidden_size=get("hidden_size", 4096),
            intermediate_size=get("intermediate_size", 11008),
            num_hidden_layers=get("num_hidden_layers", 32),
            num_attention_heads=get("num_attention_heads", 32),
            num_key_value_heads=get("num_key_value_heads", get("num_attention_heads", 32)),
            head_dim=get("head_dim", None),
            max_position_embeddings=get("max_position_embeddings", 8192),
            rms_norm_eps=get("rms_norm_eps", 1e-5),
            rope_theta=theta,
            rope_scaling=scaling,
            no_rope_layers=get("no_rope_layers", None),
            # mistral: sliding_window set => windowed; qwen2 gates it behind
            # use_sliding_window
            sliding_window=(get("sliding_window")
                            if get("use_sliding_window", True) else None),
            # ernie + helium use pair-interleaved rotate_half
            rope_interleaved=("Ernie" in archs) or ("Helium" in archs),
            tie_word_embeddings=get("tie_word_embeddings", False),
            # HF Qwen2 hardcodes qkv bias=True regardless of config fields
            attention_bias=("Qwen2ForCausalLM" in archs or "Qwen2MoeForCausalLM" in archs
                            or get("attention_bias", get("qkv_bias", False))),
            # granite multipliers (neutral when absent); granite's attention
            # scale IS the multiplier (no implicit 1/sqrt(d))
            attention_multiplier=get("attention_multiplier"),
            residual_multiplier=get("residual_multiplier", 1.0),
            embedding_multiplier=get("embedding_multiplier", 1.0),
            logits_scaling=get("logits_scaling", 1.0),
            mlp_bias=get("mlp_bias", False),
            initializer_range=get("initializer_range", 0.02),
        )


class LlamaAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig, backend: BackendConfig,
                 use_rope: bool = True):
        super().__init__()
        self.cfg = cfg
        self.backend = backend
        self.use_rope = use_rope
        H, D = cfg.num_attention_heads, cfg.head_dim
        Hk = cfg.num_key_value_heads
        self.num_heads, self.num_kv_heads, self.head_dim = H, Hk, D
        bias = cfg.attention_bias
        if cfg.fused_qkv:
            self.qkv_proj = nn.Linear(cfg.hidden_size, (H + 2 * Hk) * D, bias=bias)
        else:
            self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=bias)
            self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=bias)
            self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=bias)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        if cfg.qk_norm:
            self.q_norm = RMSNorm(D, cfg.rms_norm_eps, backend.rms_norm)
            self.k_norm = RMSNorm(D, cfg.rms_norm_eps, backend.rms_norm)
        elif cfg.qk_norm_full:
            self.q_norm = RMSNorm(H * D, cfg.rms_norm_eps, backend.rms_norm)
            self.k_norm = RMSNorm(Hk * D, cfg.rms_norm_eps, backend.rms_norm)

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        from automodel_amd.parallel.cp import active_cp, cp_flash_attention

        B, S, _ = x.shape
        if self.cfg.fused_qkv:
            qkv = self.qkv_proj(x)
            q, k, v = qkv.split([self.num_heads * self.head_dim,
                                 self.num_kv_heads * self.head_dim,
                                 self.num_kv_heads * self.head_dim], dim=-1)
            q = q.view(B, S, -1, self.head_dim)
            k = k.contiguous().view(B, S, -1, self.head_dim)
            v = v.contiguous().view(B, S, -1, self.head_dim)
        else:
            q = self.q_proj(x)
            k = self.k_proj(x)
            if self.cfg.qk_norm_full:   # OLMoE: norm over the flat projection
                q = self.q_norm(q)
                k = self.k_norm(k)
            q = q.view(B, S, -1, self.head_dim)
            k = k.view(B, S, -1, self.head_dim)
            v = self.v_proj(x).view(B, S, -1, self.head_dim)
        if self.cfg.qk_norm and not self.cfg.qk_norm_after_rope:
            q = self.q_norm(q)
            k = self.k_norm(k)
        if self.use_rope:
            if self.cfg.rope_interleaved:
                # interleaved convention: de-interleave to half-split order
                # before the standard kernel. Attention scores are invariant
                # to a head-dim permutation applied to BOTH q and k, so no
                # re-interleave is needed (v is untouched).
                d2 = self.head_dim // 2
                q = q.reshape(B, S, -1, d2, 2).transpose(-1, -2).reshape(B, S, -1, self.head_dim)
                k = k.reshape(B, S, -1, d2, 2).transpose(-1, -2).reshape(B, S, -1, self.head_dim)
            q, k = apply_rope(q, k, cos, sin, backend=self.backend.rope)
        if self.cfg.qk_norm and self.cfg.qk_norm_after_rope:
            # HunYuan-Dense: per-head q/k RMSNorm applied AFTER rope
            q = self.q_norm(q)
            k = self.k_norm(k)
        attn_scale = self.cfg.attention_multiplier  # None -> 1/sqrt(D)
        if self.cfg.sliding_window is not None:
            # windowed causal (mistral): explicit mask via sdpa — the flash
            # kernel's full-causal tiling doesn't window (round-2 kernel)
            from automodel_amd.ops.attention import sdpa_masked

            i = torch.arange(S, device=x.device)
            keep = (i[None, :] <= i[:, None]) \
                & (i[None, :] > i[:, None] - self.cfg.sliding_window)
            mask = torch.where(keep, 0.0, float("-inf")) \
                .to(q.dtype).reshape(1, 1, S, S)
            o = sdpa_masked(q, k, v, mask, scale=attn_scale)
        elif self.cfg.bidirectional:
            from automodel_amd.ops.attention import _sdpa

            o = _sdpa(q, k, v, causal=False, scale=attn_scale)
        elif active_cp() is not None:
            from automodel_amd.ops import attention as attn_mod

            if attn_mod._VARLEN_CU is not None:
                # packed documents under CP: block-diagonal-correct path
                from automodel_amd.parallel.cp import cp_blockdiag_attention

                o = cp_blockdiag_attention(q, k, v, attn_mod._VARLEN_CU,
                                           scale=attn_scale)
            else:
                o = cp_flash_attention(q, k, v, causal=True, scale=attn_scale,
                                       backend=self.backend.attn)
        else:
            from automodel_amd.utils.kv_cache import active_kv_cache, maybe_update_kv

            cache = active_kv_cache()
            if cache is not None and getattr(cache, "is_static", False):
                # static-shape decode (hipGraph capture): full-buffer KV +
                # additive mask from the device position counter
                from automodel_amd.ops.attention import sdpa_masked

                k, v, _ = maybe_update_kv(k, v)
                o = sdpa_masked(q, k, v, cache.attn_mask(), scale=attn_scale)
            else:
                k, v, cache_pos = maybe_update_kv(k, v)
                # cached decode: q is short (often 1) and prompts are
                # arbitrary lengths — sdpa is the right tool (GEMV-bound;
                # the flash kernel's 128-row tiling wants training shapes)
                attn_backend = "sdpa" if cache is not None else self.backend.attn
                o = flash_attention(q, k, v, causal=True, scale=attn_scale,
                                    backend=attn_backend, q_start=cache_pos)
        return self.o_proj(o.reshape(B, S, -1))


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig, backend: BackendConfig):
        super().__init__()
        self.backend = backend
        self.fused = cfg.fused_gate_up
        bias = cfg.mlp_bias
        if self.fused:
            self.gate_up_proj = nn.Linear(cfg.hidden_size, 2 * cfg.intermediate_size,
                                          bias=bias)
        else:
            self.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=bias)
            self.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=bias)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.fused:
            return self.down_proj(swiglu_cat(self.gate_up_proj(x)))
        return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


class XIELUActivation(nn.Module):
    """xIELU (arXiv:2411.13010, Apertus): learnable alpha_p/alpha_n stored
    in softplus-inverse form; matches HF transformers.activations math."""

    def __init__(self):
        super().__init__()
        import math as _m

        inv_sp = lambda v: _m.log(_m.expm1(v))
        self.alpha_p = nn.Parameter(torch.tensor([inv_sp(0.8)]))
        self.alpha_n = nn.Parameter(torch.tensor([inv_sp(0.8 - 0.5)]))
        self.register_buffer("beta", torch.tensor(0.5))
        self.register_buffer("eps", torch.tensor(-1e-6))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        ap = torch.nn.functional.softplus(self.alpha_p)
        an = self.beta + torch.nn.functional.softplus(self.alpha_n)
        return torch.where(
            x > 0,
            ap * x * x + self.beta * x,
            (torch.expm1(torch.min(x, self.eps.to(x.dtype))) - x) * an + self.beta * x,
        )


class XIELUMLP(nn.Module):
    """Apertus MLP: up_proj -> xIELU -> down_proj (no gate)."""

    def __init__(self, cfg: LlamaConfig, backend: BackendConfig):
        super().__init__()
        self.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=cfg.mlp_bias)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=cfg.mlp_bias)
        self.act_fn = XIELUActivation()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.down_proj(self.act_fn(self.up_proj(x)))


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig, backend: BackendConfig,
                 use_rope: bool = True):
        super().__init__()
        self.olmo2 = cfg.olmo2_layout
        self.residual_multiplier = cfg.residual_multiplier
        self.self_attn = LlamaAttention(cfg, backend, use_rope=use_rope)
        self.mlp = (XIELUMLP if cfg.mlp_type == "xielu" else LlamaMLP)(cfg, backend)
        if cfg.olmo2_layout:   # norms on outputs (OLMo-2)
            self.post_attention_layernorm = RMSNorm(cfg.hidden_size,
                                                    cfg.rms_norm_eps, backend.rms_norm)
            self.post_feedforward_layernorm = RMSNorm(cfg.hidden_size,
                                                      cfg.rms_norm_eps, backend.rms_norm)
        else:
            self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
            self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        if self.olmo2:
            x = x + self.post_attention_layernorm(self.self_attn(x, cos, sin))
            return x + self.post_feedforward_layernorm(self.mlp(x))
        r = self.residual_multiplier
        x = x + self.self_attn(self.input_layernorm(x), cos, sin) * r
        x = x + self.mlp(self.post_attention_layernorm(x)) * r
        return x


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig, backend: BackendConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        nope = cfg.no_rope_layers
        self.layers = nn.ModuleList(
            LlamaDecoderLayer(cfg, backend,
                              use_rope=(nope is None or bool(nope[i])))
            for i in range(cfg.num_hidden_layers)
        )
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        cos, sin = build_rope_cache(
            cfg.head_dim, cfg.max_position_embeddings, cfg.rope_theta, cfg.rope_scaling
        )
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, input_ids: torch.Tensor, position_ids: torch.Tensor | None = None) -> torch.Tensor:
        x = self.embed_tokens(input_ids)
        if self.cfg.embedding_multiplier != 1.0:
            x = x * self.cfg.embedding_multiplier
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = self.rope_cos[:S], self.rope_sin[:S]
        else:
            cos, sin = self.rope_cos[position_ids[0]], self.rope_sin[position_ids[0]]
        # model.to(bf16) converts buffers; the HIP rope kernel wants f32 tables
        if cos.dtype != torch.float32:
            cos, sin = cos.float(), sin.float()
        for layer in self.layers:
            x = layer(x, cos, sin)
        return self.norm(x)


class LlamaForCausalLM(nn.Module):
    hf_architectures = ("LlamaForCausalLM", "Qwen2ForCausalLM", "MistralForCausalLM",
                        "Qwen3ForCausalLM", "Phi3ForCausalLM", "SmolLM3ForCausalLM",
                        "Ernie4_5ForCausalLM", "Olmo2ForCausalLM", "GraniteForCausalLM",
                        "HeliumForCausalLM", "SeedOssForCausalLM",
                        "HunYuanDenseV1ForCausalLM", "ApertusForCausalLM")
    config_class = LlamaConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> LlamaConfig:
        return LlamaConfig.from_hf_config(hf_cfg)

    def __init__(self, config: LlamaConfig | dict, backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = LlamaConfig(**config)
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend),
            device_type, head_dim=config.head_dim,
        )
        self.config = config
        self.backend = backend
        self.model = LlamaModel(config, backend)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        if config.fused_qkv or config.fused_gate_up:
            self.state_dict_adapter = LlamaFusedStateDictAdapter(config)
        elif config.hf_key_renames:
            self.state_dict_adapter = LlamaRenameStateDictAdapter(config.hf_key_renames)
        # set by the recipe; called as loss_fn(hidden, lm_head_weight, labels).
        # Computing the loss INSIDE forward keeps lm_head.weight unsharded
        # under FSDP2 (reference computes it via lm-weight gather instead,
        # loss/linear_ce.py:147).
        self.loss_fn = None

    # -- forward paths ---------------------------------------------------------
    def forward(
        self,
        input_ids: torch.Tensor,
        labels: torch.Tensor | None = None,
        position_ids: torch.Tensor | None = None,
        return_hidden: bool = False,
        **_: Any,
    ) -> torch.Tensor:
        """labels given -> token-sum loss; return_hidden -> [B,S,H]; else logits.

        ``labels`` must already be shifted (labels[t] is the target of
        position t), matching the reference's dataset convention.
        """
        hidden = self.model(input_ids, position_ids)
        if self.config.logits_scaling != 1.0 and not return_hidden:
            # dividing hidden pre-head == dividing logits (linear head)
            hidden = hidden / self.config.logits_scaling
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before passing labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        if return_hidden:
            return hidden
        return self.lm_head(hidden)

    @torch.no_grad()
    def forward_with_aux(
        self, input_ids: torch.Tensor, aux_layers: tuple[int, ...] | None = None
    ) -> tuple[torch.Tensor, list[torch.Tensor]]:
        """-> (logits, [aux hidden [B,S,H] per requested layer]). Used by the
        EAGLE draft trainer (speculative/train_draft.py); default aux layers
        are low/mid/high like EAGLE-3's 3 auxiliary states."""
        m = self.model
        L = len(m.layers)
        if aux_layers is None:
            want = [max(0, L // 4), L // 2, max(0, L - 2)]
            picked: list[int] = []
            for w in want:  # dedupe for small L, keeping 3 distinct layers
                while w in picked and w < L - 1:
                    w += 1
                picked.append(min(w, L - 1))
            aux_layers = tuple(dict.fromkeys(picked))
        x = m.embed_tokens(input_ids)
        S = input_ids.shape[1]
        cos, sin = m.rope_cos[:S], m.rope_sin[:S]
        if cos.dtype != torch.float32:
            cos, sin = cos.float(), sin.float()
        aux = []
        for i, layer in enumerate(m.layers):
            x = layer(x, cos, sin)
            if i in aux_layers:
                aux.append(x)
        return self.lm_head(m.norm(x)), aux

    # -- init ------------------------------------------------------------------
    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        """Random init (used for synthetic benchmarks and tests)."""
        std = self.config.initializer_range

        def _init(m: nn.Module):
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, mean=0.0, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, mean=0.0, std=std)
            elif isinstance(m, RMSNorm):
                nn.init.ones_(m.weight)

        if device is not None:
            self.to_empty(device=device)
            # re-register non-persistent rope buffers lost by to_empty on meta
            cos, sin = build_rope_cache(
                self.config.head_dim,
                self.config.max_position_embeddings,
                self.config.rope_theta,
                self.config.rope_scaling,
                device=device,
            )
            self.model.rope_cos.copy_(cos)
            self.model.rope_sin.copy_(sin)
        self.apply(_init)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        seen, total = set(), 0
        for p in self.parameters():
            if id(p) not in seen:
                seen.add(id(p))
                total += p.numel()
        return total


class LlamaRenameStateDictAdapter:
    """Substring-rename adapter for llama-family archs whose HF key names
    differ (HunYuan-Dense norm names, Apertus pre-norm names)."""

    def __init__(self, renames: dict):
        self.renames = dict(renames)

    def _apply(self, sd: dict, mapping: dict) -> dict:
        out = {}
        for k, v in sd.items():
            for a, b in mapping.items():
                if a in k:
                    k = k.replace(a, b)
            out[k] = v
        return out

    def from_hf(self, sd: dict) -> dict:
        return self._apply(sd, self.renames)

    def to_hf(self, sd: dict) -> dict:
        return self._apply(sd, {b: a for a, b in self.renames.items()})

    def hf_key_targets(self, key: str) -> list:
        k = key
        for a, b in self.renames.items():
            if a in k:
                k = k.replace(a, b)
        return [k]


class LlamaFusedStateDictAdapter:
    """HF per-projection keys <-> fused qkv / gate_up weights."""

    def __init__(self, config: LlamaConfig):
        self.cfg = config

    def hf_key_targets(self, key: str) -> list[str]:
        """Internal param name(s) an HF key feeds (streaming HF loader)."""
        import re

        m = re.match(r"^(model\.layers\.\d+)\.self_attn\.[qkv]_proj\.(weight|bias)$", key)
        if m and self.cfg.fused_qkv:
            return [f"{m.group(1)}.self_attn.qkv_proj.{m.group(2)}"]
        m = re.match(r"^(model\.layers\.\d+)\.mlp\.(?:gate|up)_proj\.weight$", key)
        if m and self.cfg.fused_gate_up:
            return [f"{m.group(1)}.mlp.gate_up_proj.weight"]
        return [key]

    def from_hf(self, sd: dict) -> dict:
        import re

        cfg = self.cfg
        out = dict(sd)
        for i in range(cfg.num_hidden_layers):
            p = f"model.layers.{i}"
            if cfg.fused_qkv:
                parts = [out.pop(f"{p}.self_attn.{n}_proj.weight", None) for n in "qkv"]
                if all(x is not None for x in parts):
                    out[f"{p}.self_attn.qkv_proj.weight"] = torch.cat(parts, dim=0)
                bparts = [out.pop(f"{p}.self_attn.{n}_proj.bias", None) for n in "qkv"]
                if all(x is not None for x in bparts):
                    out[f"{p}.self_attn.qkv_proj.bias"] = torch.cat(bparts, dim=0)
            if cfg.fused_gate_up:
                g = out.pop(f"{p}.mlp.gate_proj.weight", None)
                u = out.pop(f"{p}.mlp.up_proj.weight", None)
                if g is not None and u is not None:
                    out[f"{p}.mlp.gate_up_proj.weight"] = torch.cat([g, u], dim=0)
        return out

    def to_hf(self, sd: dict) -> dict:
        cfg = self.cfg
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        out = {}
        for k, t in sd.items():
            if k.endswith(".self_attn.qkv_proj.weight") or k.endswith(".self_attn.qkv_proj.bias"):
                q, kk, v = t.split([H * D, Hk * D, Hk * D], dim=0)
                kind = "weight" if k.endswith("weight") else "bias"
                base = k.rsplit(".qkv_proj.", 1)[0]
                out[f"{base}.q_proj.{kind}"] = q
                out[f"{base}.k_proj.{kind}"] = kk
                out[f"{base}.v_proj.{kind}"] = v
            elif k.endswith(".mlp.gate_up_proj.weight"):
                g, u = t.chunk(2, dim=0)
                base = k.rsplit(".gate_up_proj.", 1)[0]
                out[f"{base}.gate_proj.weight"] = g
                out[f"{base}.up_proj.weight"] = u
            else:
                out[k] = t
        return out
