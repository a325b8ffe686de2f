"""Kimi-Linear (KDA) hybrid LM, MI355X-native.

Reference behavior: nemo_automodel's kimi_linear family (components/
models/kimi_linear/, ~2.3k LoC). Implemented against the public
Kimi-Linear architecture (Kimi Delta Attention):

  * KDA mixer: the gated delta rule with PER-CHANNEL decay — state
    ``S_t = (I - b_t k_t k_t^T) Diag(a_t) S_{t-1} + b_t k_t v_t^T`` with
    ``a_t in (0,1)^{d_k}`` (GatedDeltaNet's scalar alpha becomes a
    channelwise diagonal), l2-normalized q/k, a grouped short conv on
    q/k/v, a low-rank decay projection, and a low-rank sigmoid output
    gate (the attention-sink fix);
  * hybrid schedule: every ``full_attn_interval``-th layer is full MLA
    attention with NO position encoding (NoPE — identity rope tables),
    the rest are KDA;
  * MoE FFNs (sigmoid routing + shared expert, dense-first) via the
    in-tree MoE stack.

``kda_chunked`` carries the chunk math: decays folded into transformed
keys (exponent differences are <= 0 inside a chunk, so every exp is
bounded), the per-chunk unit-triangular system solved with ONE batched
triangular solve; the O(c^2 d_k) pairwise decay tensor is the torch
reference the (future) HIP block-tiled kernel is tested against.
``kda_recurrent`` is the exact per-token loop used for numerics tests.

No kimi_linear exists in this image's transformers, so tests are
internal: chunked == recurrent, and scalar-decay reduction matches the
qwen3_next gated delta rule bit-for-bit.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.deepseek_v3.model import DeepseekV3Config, DenseMLP, MLAAttention
from automodel_amd.models.qwen3_next.model import GatedHeadNorm, _l2norm
from automodel_amd.moe.config import MoEConfig
from automodel_amd.moe.layers import MoE
from automodel_amd.ops.rms_norm import RMSNorm


def kda_recurrent(q, k, v, g, beta):
    """Exact per-token KDA reference. q/k [B,S,H,Dk]; v [B,S,H,Dv];
    g [B,S,H,Dk] per-channel LOG decay (<=0); beta [B,S,H]."""
    q = _l2norm(q.float()) * (q.shape[-1] ** -0.5)
    k = _l2norm(k.float())
    v = v.float()
    B, S, H, Dk = k.shape
    Dv = v.shape[-1]
    state = q.new_zeros(B, H, Dk, Dv)
    outs = []
    for t in range(S):
        state = state * g[:, t].float().exp().unsqueeze(-1)   # Diag(a_t)
        kt = k[:, t]                                          # [B,H,Dk]
        bt = beta[:, t].float()
        err = torch.einsum("bhd,bhdv->bhv", kt, state)        # k^T Diag(a) S
        state = state - bt[..., None, None] * kt.unsqueeze(-1) * err.unsqueeze(-2)
        state = state + bt[..., None, None] * kt.unsqueeze(-1) * v[:, t].float().unsqueeze(-2)
        outs.append(torch.einsum("bhd,bhdv->bhv", q[:, t], state))
    return torch.stack(outs, dim=1)                           # [B,S,H,Dv]


def kda_chunked(q, k, v, g, beta, chunk_size: int = 16,
                initial_state=None, return_final_state=False):
    """Chunked KDA (float32): per-channel-decay generalization of the
    qwen3_next chunked gated delta rule. Same I/O as ``kda_recurrent``;
    initial/final state support the CP chunk relay."""
    q = (_l2norm(q.transpose(1, 2).float()))
    k = _l2norm(k.transpose(1, 2).float())
    v = v.transpose(1, 2).float()
    beta = beta.transpose(1, 2).float()
    g = g.transpose(1, 2).float()                # [B,H,S,Dk]
    b, h, s, dk = k.shape
    dv = v.shape[-1]
    pad = (chunk_size - s % chunk_size) % chunk_size
    if pad:
        q = F.pad(q, (0, 0, 0, pad))
        k = F.pad(k, (0, 0, 0, pad))
        v = F.pad(v, (0, 0, 0, pad))
        beta = F.pad(beta, (0, pad))
        g = F.pad(g, (0, 0, 0, pad))
    n = (s + pad) // chunk_size
    c = chunk_size
    q = q.view(b, h, n, c, dk) * (dk ** -0.5)
    k = k.view(b, h, n, c, dk)
    v = v.view(b, h, n, c, dv)
    beta = beta.view(b, h, n, c)
    G = g.view(b, h, n, c, dk).cumsum(dim=-2)    # per-channel log cumdecay

    v_beta = v * beta.unsqueeze(-1)
    k_beta = k * beta.unsqueeze(-1)
    state = (q.new_zeros(b, h, dk, dv) if initial_state is None
             else initial_state.to(q.dtype))
    out = torch.empty(b, h, n, c, dv, dtype=q.dtype, device=q.device)
    low = torch.ones(c, c, dtype=torch.bool, device=q.device).tril()
    for i in range(n):
        Gi = G[:, :, i]                                          # [b,h,c,dk]
        # pairwise per-channel decay  E[t,s,d] = exp(G_t - G_s), t >= s
        E = (Gi.unsqueeze(-2) - Gi.unsqueeze(-3)).clamp(max=0).exp()
        a0 = -torch.einsum("bhtd,bhsd,bhtsd->bhts",
                           k_beta[:, :, i], k[:, :, i], E).tril(-1)
        eye = torch.eye(c, dtype=a0.dtype, device=a0.device).expand_as(a0)
        T = torch.linalg.solve_triangular(eye - a0, eye.contiguous(),
                                          upper=False, unitriangular=True)
        v_in = T @ v_beta[:, :, i]
        k_cumdecay = T @ (k_beta[:, :, i] * Gi.exp())
        attn = torch.einsum("bhtd,bhsd,bhtsd->bhts",
                            q[:, :, i], k[:, :, i], E).masked_fill(~low, 0)
        v_prime = k_cumdecay @ state
        v_new = v_in - v_prime
        out[:, :, i] = (q[:, :, i] * Gi.exp()) @ state + attn @ v_new
        glast = Gi[:, :, -1]                                     # [b,h,dk]
        state = (state * glast.exp().unsqueeze(-1) +
                 (k[:, :, i] * (glast.unsqueeze(-2) - Gi).exp()
                  ).transpose(-1, -2) @ v_new)
    out = out.reshape(b, h, -1, dv)[:, :, :s].transpose(1, 2)
    if return_final_state:
        return out, state
    return out


@dataclass
class KimiLinearConfig:
    vocab_size: int = 163840
    hidden_size: int = 2048
    intermediate_size: int = 11264
    num_hidden_layers: int = 27
    rms_norm_eps: float = 1e-5
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02
    # ---- KDA mixer
    linear_num_heads: int = 32
    linear_head_dim: int = 128
    linear_conv_kernel: int = 4
    linear_lowrank: int = 32
    full_attn_interval: int = 4        # every 4th layer is full MLA (NoPE)
    # ---- MLA (full-attention layers)
    num_attention_heads: int = 16
    q_lora_rank: int = 0
    kv_lora_rank: int = 512
    qk_nope_head_dim: int = 128
    qk_rope_head_dim: int = 64         # NoPE: rope table is identity
    v_head_dim: int = 128
    # ---- FFN
    first_k_dense_replace: int = 1
    moe: MoEConfig = field(default_factory=lambda: MoEConfig(
        n_routed_experts=8, n_shared_experts=1, n_activated_experts=2,
        score_func="sigmoid", expert_bias=True, norm_topk_prob=True,
        moe_intermediate_size=1024, shared_expert_intermediate_size=1024))
    max_position_embeddings: int = 131072

    def __post_init__(self):
        if isinstance(self.moe, dict):
            self.moe = MoEConfig(**self.moe)

    @property
    def head_dim(self):
        return self.qk_nope_head_dim + self.qk_rope_head_dim

    @classmethod
    def from_hf_config(cls, hf: Any) -> "KimiLinearConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        la = hf.get("linear_attn_config", hf)
        return cls(
            vocab_size=g("vocab_size", 163840),
            hidden_size=g("hidden_size", 2048),
            intermediate_size=g("intermediate_size", 11264),
            num_hidden_layers=g("num_hidden_layers", 27),
            rms_norm_eps=g("rms_norm_eps", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", False),
            linear_num_heads=la.get("num_heads", g("linear_num_heads", 32)),
            linear_head_dim=la.get("head_dim", g("linear_head_dim", 128)),
            linear_conv_kernel=la.get("short_conv_kernel_size",
                                      g("linear_conv_kernel", 4)),
            full_attn_interval=la.get("full_attn_layers",
                                      g("full_attn_interval", 4)),
            num_attention_heads=g("num_attention_heads", 16),
            q_lora_rank=g("q_lora_rank") or 0,
            kv_lora_rank=g("kv_lora_rank", 512),
            qk_nope_head_dim=g("qk_nope_head_dim", 128),
            qk_rope_head_dim=g("qk_rope_head_dim", 64),
            v_head_dim=g("v_head_dim", 128),
            first_k_dense_replace=g("first_k_dense_replace", 1),
            moe=MoEConfig(
                n_routed_experts=g("n_routed_experts", 8) or 8,
                n_shared_experts=g("n_shared_experts", 1),
                n_activated_experts=g("num_experts_per_tok", 2),
                score_func="sigmoid", expert_bias=True,
                norm_topk_prob=g("norm_topk_prob", True),
                route_scale=g("routed_scaling_factor", 1.0),
                moe_intermediate_size=g("moe_intermediate_size", 1024),
                shared_expert_intermediate_size=(
                    g("moe_intermediate_size", 1024)
                    * g("n_shared_experts", 1)),
            ),
            max_position_embeddings=g("max_position_embeddings", 131072),
        )


class KimiDeltaAttention(nn.Module):
    """KDA mixer: short conv + per-channel-decay delta rule + output gate."""

    def __init__(self, cfg: KimiLinearConfig):
        super().__init__()
        H, D = cfg.linear_num_heads, cfg.linear_head_dim
        hid = cfg.hidden_size
        self.n_heads, self.head_dim = H, D
        self.q_proj = nn.Linear(hid, H * D, bias=False)
        self.k_proj = nn.Linear(hid, H * D, bias=False)
        self.v_proj = nn.Linear(hid, H * D, bias=False)
        self.conv_dim = 3 * H * D
        self.conv1d = nn.Conv1d(self.conv_dim, self.conv_dim,
                                cfg.linear_conv_kernel, groups=self.conv_dim,
                                padding=cfg.linear_conv_kernel - 1, bias=False)
        # low-rank per-channel decay:  a = -exp(A_log) * softplus(f2(f1(x)) + dt_bias)
        r = cfg.linear_lowrank
        self.f_a_proj = nn.Linear(hid, r, bias=False)
        self.f_b_proj = nn.Linear(r, H * D, bias=False)
        self.dt_bias = nn.Parameter(torch.ones(H * D))
        self.A_log = nn.Parameter(torch.zeros(H))
        self.b_proj = nn.Linear(hid, H, bias=False)           # beta
        # low-rank sigmoid output gate (attention-sink fix)
        self.g_a_proj = nn.Linear(hid, r, bias=False)
        self.g_b_proj = nn.Linear(r, H * D, bias=False)
        self.o_norm = GatedHeadNorm(D, eps=1e-5)
        self.o_proj = nn.Linear(H * D, hid, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, _ = x.shape
        H, D = self.n_heads, self.head_dim
        qkv = torch.cat([self.q_proj(x), self.k_proj(x), self.v_proj(x)], dim=-1)
        qkv = F.silu(self.conv1d(qkv.transpose(1, 2))[..., :S].transpose(1, 2))
        q, k, v = qkv.split([H * D, H * D, H * D], dim=-1)
        q = q.view(B, S, H, D)
        k = k.view(B, S, H, D)
        v = v.view(B, S, H, D)
        g = (-self.A_log.float().exp().repeat_interleave(D)
             * F.softplus(self.f_b_proj(self.f_a_proj(x)).float() + self.dt_bias)
             ).view(B, S, H, D)
        beta = self.b_proj(x).sigmoid()
        o = kda_chunked(q, k, v, g, beta).to(x.dtype)         # [B,S,H,D]
        gate = self.g_b_proj(self.g_a_proj(x)).view(-1, D)
        o = self.o_norm(o.reshape(-1, D), gate)
        return self.o_proj(o.view(B, S, H * D))


class KimiLinearDecoderLayer(nn.Module):
    def __init__(self, cfg: KimiLinearConfig, backend: BackendConfig,
                 layer_idx: int):
        super().__init__()
        self.is_full = (layer_idx + 1) % cfg.full_attn_interval == 0
        if self.is_full:
            mla_cfg = DeepseekV3Config(
                hidden_size=cfg.hidden_size,
                num_attention_heads=cfg.num_attention_heads,
                q_lora_rank=cfg.q_lora_rank, kv_lora_rank=cfg.kv_lora_rank,
                qk_nope_head_dim=cfg.qk_nope_head_dim,
                qk_rope_head_dim=cfg.qk_rope_head_dim,
                v_head_dim=cfg.v_head_dim, rms_norm_eps=cfg.rms_norm_eps)
            self.self_attn = MLAAttention(mla_cfg, backend)
        else:
            self.self_attn = KimiDeltaAttention(cfg)
        if layer_idx < cfg.first_k_dense_replace:
            self.mlp = DenseMLP(cfg.hidden_size, cfg.intermediate_size)
        else:
            self.mlp = MoE(cfg.hidden_size, cfg.moe)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                       backend.rms_norm)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size,
                                                cfg.rms_norm_eps,
                                                backend.rms_norm)

    def forward(self, x, cos, sin):
        h = self.input_layernorm(x)
        x = x + (self.self_attn(h, cos, sin) if self.is_full
                 else self.self_attn(h))
        return x + self.mlp(self.post_attention_layernorm(x))


class KimiLinearForCausalLM(nn.Module):
    hf_architectures = ("KimiLinearForCausalLM",)
    config_class = KimiLinearConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> KimiLinearConfig:
        return KimiLinearConfig.from_hf_config(hf_cfg)

    def __init__(self, config: KimiLinearConfig | dict,
                 backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = KimiLinearConfig(**config)
        self.config = config
        bk = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend),
            "cuda" if torch.cuda.is_available() else "cpu",
            head_dim=config.head_dim)
        self.backend = bk
        inner = nn.Module()
        inner.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        inner.layers = nn.ModuleList(
            KimiLinearDecoderLayer(config, bk, i)
            for i in range(config.num_hidden_layers))
        inner.norm = RMSNorm(config.hidden_size, config.rms_norm_eps,
                             bk.rms_norm)
        # NoPE: MLA layers run with identity rope tables
        S = config.max_position_embeddings
        inner.register_buffer("rope_cos",
                              torch.ones(S, config.qk_rope_head_dim),
                              persistent=False)
        inner.register_buffer("rope_sin",
                              torch.zeros(S, config.qk_rope_head_dim),
                              persistent=False)
        self.model = inner
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size,
                                 bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = inner.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None,
                return_hidden=False, **_: Any):
        m = self.model
        x = m.embed_tokens(input_ids)
        S = input_ids.shape[1]
        cos, sin = m.rope_cos[:S].float(), m.rope_sin[:S].float()
        for layer in m.layers:
            x = layer(x, cos, sin)
        hidden = m.norm(x)
        if return_hidden:
            return hidden
        if labels is not None and self.loss_fn is not None:
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        logits = self.lm_head(hidden)
        if labels is not None:
            return F.cross_entropy(
                logits.reshape(-1, logits.shape[-1]).float(),
                labels.reshape(-1), ignore_index=-100, reduction="sum")
        return logits

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            self.model.rope_cos.fill_(1.0)
            self.model.rope_sin.zero_()
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding, nn.Conv1d)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, RMSNorm) or type(mod).__name__ in (
                    "RMSNorm", "GatedHeadNorm"):
                nn.init.ones_(mod.weight)
        for mod in self.modules():
            if isinstance(mod, KimiDeltaAttention):
                mod.dt_bias.fill_(1.0)
                mod.A_log.zero_()
            elif isinstance(mod, MoE):
                mod.experts.init_weights(std)
                nn.init.normal_(mod.gate.weight, std=std)
                if getattr(mod.gate, "e_score_correction_bias", None) is not None:
                    mod.gate.e_score_correction_bias.zero_()
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
