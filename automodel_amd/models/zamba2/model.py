"""Zamba2 (Zyphra shared-transformer-block hybrid Mamba2) causal LM.

Reference behavior: the public Zamba2 architecture (HF
transformers.models.zamba2) — a Mamba2 backbone where every "hybrid" layer
routes through one of ``num_mem_blocks`` SHARED transformer blocks
(attention over concat(hidden, original_embeds) at 2H width, scale
(head_dim/2)^-0.5, then a tied MLP with per-use LoRA-style gate_up
adapters), projected by a per-layer ``linear`` and injected into that
layer's Mamba decoder input. Shared blocks are registered under every
using layer (state-dict paths mirror HF; the tensors are the same
objects). Reuses the shared chunked-SSD Mamba2Mixer (group-wise gated
norm = Zamba2RMSNormGated semantics).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.nemotron_h.model import Mamba2Mixer
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.rope import apply_rope, build_rope_cache
from automodel_amd.ops.swiglu import swiglu


@dataclass
class Zamba2Config:
    vocab_size: int = 32000
    hidden_size: int = 2560
    intermediate_size: int = 4096
    num_hidden_layers: int = 54
    layers_block_type: list = field(default_factory=list)
    num_attention_heads: int = 32
    num_key_value_heads: int = 32
    attention_head_dim: int | None = None
    mamba_n_heads: int | None = None
    mamba_headdim: int = 64
    mamba_d_state: int = 64
    mamba_ngroups: int = 1
    mamba_d_conv: int = 4
    mamba_expand: int = 2
    num_mem_blocks: int = 1
    adapter_rank: int = 128
    use_shared_attention_adapter: bool = False
    use_mem_rope: bool = False
    rope_theta: float = 10000.0
    add_bias_linear: bool = False
    hidden_act: str = "gelu"
    time_step_min: float = 0.001
    rms_norm_eps: float = 1e-5
    max_position_embeddings: int = 4096
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    def __post_init__(self):
        if not self.layers_block_type:
            self.layers_block_type = ["mamba"] * self.num_hidden_layers
        if self.attention_head_dim is None:
            self.attention_head_dim = 2 * self.hidden_size // self.num_attention_heads
        if self.mamba_n_heads is None:
            self.mamba_n_heads = (self.mamba_expand * self.hidden_size
                                  ) // self.mamba_headdim

    @property
    def hybrid_layer_ids(self) -> list[int]:
        return [i for i, t in enumerate(self.layers_block_type) if t == "hybrid"]

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Zamba2Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        return cls(
            vocab_size=g("vocab_size", 32000),
            hidden_size=g("hidden_size", 2560),
            intermediate_size=g("intermediate_size", 4096),
            num_hidden_layers=g("num_hidden_layers", 54),
            layers_block_type=g("layers_block_type") or [],
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads") or g("num_attention_heads", 32),
            attention_head_dim=g("attention_head_dim"),
            mamba_n_heads=g("mamba_n_heads") or g("n_mamba_heads"),
            mamba_headdim=g("mamba_headdim", 64),
            mamba_d_state=g("mamba_d_state", 64),
            mamba_ngroups=g("mamba_ngroups", 1),
            mamba_d_conv=g("mamba_d_conv", 4),
            mamba_expand=g("mamba_expand", 2),
            num_mem_blocks=g("num_mem_blocks", 1),
            adapter_rank=g("adapter_rank", 128),
            use_shared_attention_adapter=g("use_shared_attention_adapter", False),
            use_mem_rope=g("use_mem_rope", False),
            rope_theta=g("rope_theta") or 10000.0,
            add_bias_linear=g("add_bias_linear", False),
            hidden_act=g("hidden_act", "gelu"),
            time_step_min=g("time_step_min", 0.001),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            max_position_embeddings=g("max_position_embeddings", 4096),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


def _adapter(in_dim: int, rank: int, out_dim: int) -> nn.Sequential:
    return nn.Sequential(nn.Linear(in_dim, rank, bias=False),
                         nn.Linear(rank, out_dim, bias=False))


class Zamba2Attention(nn.Module):
    """Attention at 2H width over concat(hidden, embeds); optional per-use
    q/k/v adapters; scale (head_dim/2)^-0.5."""

    def __init__(self, cfg: Zamba2Config, block_id: int, backend: BackendConfig):
        super().__init__()
        ah = 2 * cfg.hidden_size
        D = cfg.attention_head_dim
        H, Hk = cfg.num_attention_heads, cfg.num_key_value_heads
        self.head_dim = D
        self.scale = (D / 2) ** -0.5
        self.use_rope = cfg.use_mem_rope
        self.q_proj = nn.Linear(ah, H * D, bias=False)
        self.k_proj = nn.Linear(ah, Hk * D, bias=False)
        self.v_proj = nn.Linear(ah, Hk * D, bias=False)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        self.use_adapter = cfg.use_shared_attention_adapter
        if self.use_adapter:
            n = len(cfg.hybrid_layer_ids)
            mk = (lambda i: _adapter(ah, cfg.adapter_rank, ah)
                  if i % cfg.num_mem_blocks == block_id else nn.Identity())
            self.linear_q_adapter_list = nn.ModuleList(mk(i) for i in range(n))
            self.linear_k_adapter_list = nn.ModuleList(mk(i) for i in range(n))
            self.linear_v_adapter_list = nn.ModuleList(mk(i) for i in range(n))
        self.backend = backend

    def forward(self, h, use_idx: int, cos, sin):
        B, S, _ = h.shape
        D = self.head_dim
        q, k, v = self.q_proj(h), self.k_proj(h), self.v_proj(h)
        if self.use_adapter:
            q = q + self.linear_q_adapter_list[use_idx](h)
            k = k + self.linear_k_adapter_list[use_idx](h)
            v = v + self.linear_v_adapter_list[use_idx](h)
        q = q.view(B, S, -1, D)
        k = k.view(B, S, -1, D)
        v = v.view(B, S, -1, D)
        if self.use_rope:
            q, k = apply_rope(q, k, cos, sin, backend="torch")
        o = flash_attention(q, k, v, causal=True, scale=self.scale,
                            backend=self.backend.attn)
        return self.o_proj(o.reshape(B, S, -1))


class Zamba2MLP(nn.Module):
    """Tied gate_up/down with per-use LoRA-style gate_up adapters."""

    def __init__(self, cfg: Zamba2Config, block_id: int):
        super().__init__()
        b = cfg.add_bias_linear
        self.gate_up_proj = nn.Linear(cfg.hidden_size, 2 * cfg.intermediate_size, bias=b)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=b)
        n = len(cfg.hybrid_layer_ids)
        self.gate_up_proj_adapter_list = nn.ModuleList(
            _adapter(cfg.hidden_size, cfg.adapter_rank, 2 * cfg.intermediate_size)
            if i % cfg.num_mem_blocks == block_id else nn.Identity()
            for i in range(n))
        self.hidden_act = cfg.hidden_act    # Zamba2 default is GELU-gated

    def forward(self, x, use_idx: int):
        gu = self.gate_up_proj(x) + self.gate_up_proj_adapter_list[use_idx](x)
        gate, up = gu.chunk(2, dim=-1)
        if self.hidden_act == "silu":
            return self.down_proj(swiglu(gate, up))
        act = getattr(torch.nn.functional, self.hidden_act)
        return self.down_proj(act(gate) * up)


class Zamba2SharedBlock(nn.Module):
    """One tied transformer block (attention + MLP, NO internal residuals)."""

    def __init__(self, cfg: Zamba2Config, block_id: int, backend: BackendConfig):
        super().__init__()
        self.self_attn_block_id = block_id
        self.self_attn = Zamba2Attention(cfg, block_id, backend)
        self.feed_forward = Zamba2MLP(cfg, block_id)
        self.input_layernorm = RMSNorm(2 * cfg.hidden_size, cfg.rms_norm_eps,
                                       backend.rms_norm)
        self.pre_ff_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                        backend.rms_norm)

    def forward(self, x, x0, use_idx: int, cos, sin):
        h = self.input_layernorm(torch.cat([x, x0], dim=-1))
        h = self.self_attn(h, use_idx, cos, sin)
        return self.feed_forward(self.pre_ff_layernorm(h), use_idx)


class Zamba2MambaDecoder(nn.Module):
    def __init__(self, cfg: Zamba2Config, backend: BackendConfig):
        super().__init__()
        inter = cfg.mamba_expand * cfg.hidden_size
        self.mamba = Mamba2Mixer(
            cfg.hidden_size, cfg.mamba_n_heads, cfg.mamba_headdim,
            cfg.mamba_d_state, cfg.mamba_ngroups, cfg.mamba_d_conv,
            chunk_size=128, eps=cfg.rms_norm_eps,
            use_bias=False, use_conv_bias=True,
            norm_group_size=inter // cfg.mamba_ngroups,
            time_step_limit=(cfg.time_step_min, float("inf")))
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                       backend.rms_norm)

    def forward(self, x, extra=None):
        r = x
        if extra is not None:
            x = x + extra
        return r + self.mamba(self.input_layernorm(x))


class Zamba2HybridLayer(nn.Module):
    def __init__(self, cfg: Zamba2Config, shared: Zamba2SharedBlock,
                 use_idx: int, backend: BackendConfig):
        super().__init__()
        self.shared_transformer = shared        # SHARED module object
        self.linear = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=False)
        self.mamba_decoder = Zamba2MambaDecoder(cfg, backend)
        self.use_idx = use_idx

    def forward(self, x, x0, cos, sin):
        t = self.linear(self.shared_transformer(x, x0, self.use_idx, cos, sin))
        return self.mamba_decoder(x, extra=t)


class Zamba2Model(nn.Module):
    def __init__(self, cfg: Zamba2Config, backend: BackendConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        # HF parity: a DISTINCT block per hybrid layer at init (block_id =
        # global layer % num_mem_blocks decides which adapter slots are
        # real); true weight sharing is applied by tie_shared_blocks() after
        # loading a real checkpoint — exactly the reference's
        # _tied_weights_keys behavior (modeling_zamba2.py get_layers).
        layers = []
        j = 0
        for gi, t in enumerate(cfg.layers_block_type):
            if t == "hybrid":
                block = Zamba2SharedBlock(cfg, gi % cfg.num_mem_blocks, backend)
                layers.append(Zamba2HybridLayer(cfg, block, j, backend))
                j += 1
            else:
                layers.append(Zamba2MambaDecoder(cfg, backend))
        self.layers = nn.ModuleList(layers)
        self.final_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                       backend.rms_norm)
        if cfg.use_mem_rope:
            cos, sin = build_rope_cache(cfg.attention_head_dim,
                                        min(cfg.max_position_embeddings, 32768),
                                        cfg.rope_theta)
            self.register_buffer("rope_cos", cos, persistent=False)
            self.register_buffer("rope_sin", sin, persistent=False)
        else:
            self.rope_cos = self.rope_sin = None

    def forward(self, ids):
        x = self.embed_tokens(ids)
        x0 = x
        S = x.shape[1]
        if self.rope_cos is not None:
            cos, sin = self.rope_cos[:S].float(), self.rope_sin[:S].float()
        else:
            cos = sin = None
        for layer in self.layers:
            if isinstance(layer, Zamba2HybridLayer):
                x = layer(x, x0, cos, sin)
            else:
                x = layer(x)
        return self.final_layernorm(x)


class Zamba2ForCausalLM(nn.Module):
    hf_architectures = ("Zamba2ForCausalLM",)
    config_class = Zamba2Config

    @staticmethod
    def config_from_hf(hf_cfg) -> Zamba2Config:
        return Zamba2Config.from_hf_config(hf_cfg)

    def __init__(self, config: Zamba2Config | dict, backend=None):
        super().__init__()
        cfg = config if isinstance(config, Zamba2Config) else Zamba2Config(**dict(config))
        self.config = cfg
        bk = BackendConfig.resolve(backend,
                                   "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.attention_head_dim)
        self.model = Zamba2Model(cfg, bk)
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
            if cfg.use_mem_rope:
                cos, sin = build_rope_cache(cfg.attention_head_dim,
                                            min(cfg.max_position_embeddings, 32768),
                                            cfg.rope_theta, device=device)
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
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def tie_shared_blocks(self) -> int:
        """Point every hybrid layer's shared_transformer at the FIRST block
        with the same block_id (true Zamba2 weight tying — the reference
        applies this on from_pretrained via _tied_weights_keys; checkpoints
        store only the first occurrence). Adapters stay per-layer: each
        use's real adapter lives on the canonical block's ModuleList at the
        use index, so tying merges them correctly. Returns ties applied."""
        canonical: dict[int, Zamba2SharedBlock] = {}
        n = 0
        for layer in self.model.layers:
            if not isinstance(layer, Zamba2HybridLayer):
                continue
            bid = layer.shared_transformer.self_attn_block_id
            if bid not in canonical:
                canonical[bid] = layer.shared_transformer
            elif layer.shared_transformer is not canonical[bid]:
                # graft this use's real adapters onto the canonical block
                src = layer.shared_transformer
                dst = canonical[bid]
                for i, a in enumerate(src.feed_forward.gate_up_proj_adapter_list):
                    if not isinstance(a, nn.Identity):
                        dst.feed_forward.gate_up_proj_adapter_list[i] = a
                if src.self_attn.use_adapter:
                    for lname in ("linear_q_adapter_list", "linear_k_adapter_list",
                                  "linear_v_adapter_list"):
                        sl = getattr(src.self_attn, lname)
                        dl = getattr(dst.self_attn, lname)
                        for i, a in enumerate(sl):
                            if not isinstance(a, nn.Identity):
                                dl[i] = a
                layer.shared_transformer = dst
                n += 1
        return n

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
