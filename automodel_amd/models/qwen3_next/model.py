"""Qwen3-Next (hybrid GatedDeltaNet linear attention + gated GQA + MoE),
MI355X-native.

Reference behavior: the public Qwen3-Next architecture (HF
transformers.models.qwen3_next) — layer_types mix of "linear_attention"
(GatedDeltaNet: short conv over qkv, sigmoid-beta delta rule with
log-decay gates, per-head gated RMSNorm) and "full_attention" (GQA with a
sigmoid output gate carved from q_proj, zero-centered q/k norms, partial
rotary 0.25), MoE blocks with softmax-then-topk routing plus a
sigmoid-gated shared expert. All norms except the deltanet gated norm are
zero-centered ((1+w) scaling).

The chunked delta rule here replaces the reference's per-row forward-
substitution loop with one batched `torch.linalg.solve_triangular` per
chunk block (numerically identical — the loop IS forward substitution),
keeping the whole scan GEMM-shaped for MFMA execution on MI355X; the
inter-chunk recurrence stays a seq/chunk-length python loop over batched
GEMMs (rocBLAS strided-batched).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.moe.config import MoEConfig
from automodel_amd.moe.layers import MoE
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rope import apply_rope, build_rope_cache
from automodel_amd.ops.swiglu import swiglu


@dataclass
class Qwen3NextConfig:
    vocab_size: int = 151936
    hidden_size: int = 2048
    intermediate_size: int = 5120
    num_hidden_layers: int = 48
    num_attention_heads: int = 16
    num_key_value_heads: int = 2
    head_dim: int = 256
    layer_types: list = field(default_factory=list)
    # linear attention (GatedDeltaNet)
    linear_num_value_heads: int = 32
    linear_num_key_heads: int = 16
    linear_key_head_dim: int = 128
    linear_value_head_dim: int = 128
    linear_conv_kernel_dim: int = 4
    # MoE
    num_experts: int = 512
    num_experts_per_tok: int = 10
    moe_intermediate_size: int = 512
    shared_expert_intermediate_size: int = 512
    decoder_sparse_step: int = 1
    mlp_only_layers: list = field(default_factory=list)
    norm_topk_prob: bool = True
    router_aux_loss_coef: float = 0.0
    rms_norm_eps: float = 1e-6
    rope_theta: float = 10000.0
    rope_scaling: dict | None = None
    partial_rotary_factor: float = 0.25
    attention_bias: bool = False
    max_position_embeddings: int = 262144
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Qwen3NextConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 151936),
            hidden_size=g("hidden_size", 2048),
            intermediate_size=g("intermediate_size", 5120),
            num_hidden_layers=g("num_hidden_layers", 48),
            num_attention_heads=g("num_attention_heads", 16),
            num_key_value_heads=g("num_key_value_heads", 2),
            head_dim=g("head_dim") or g("hidden_size", 2048) // g("num_attention_heads", 16),
            layer_types=g("layer_types") or [],
            linear_num_value_heads=g("linear_num_value_heads", 32),
            linear_num_key_heads=g("linear_num_key_heads", 16),
            linear_key_head_dim=g("linear_key_head_dim", 128),
            linear_value_head_dim=g("linear_value_head_dim", 128),
            linear_conv_kernel_dim=g("linear_conv_kernel_dim", 4),
            num_experts=g("num_experts", 512),
            num_experts_per_tok=g("num_experts_per_tok", 10),
            moe_intermediate_size=g("moe_intermediate_size", 512),
            shared_expert_intermediate_size=g("shared_expert_intermediate_size", 512),
            decoder_sparse_step=g("decoder_sparse_step", 1),
            mlp_only_layers=g("mlp_only_layers") or [],
            norm_topk_prob=g("norm_topk_prob", True),
            router_aux_loss_coef=g("router_aux_loss_coef", 0.0),
            rms_norm_eps=g("rms_norm_eps", 1e-6),
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            rope_scaling=g("rope_scaling"),
            partial_rotary_factor=rp.get("partial_rotary_factor",
                                         g("partial_rotary_factor", 0.25)),
            attention_bias=g("attention_bias", False),
            max_position_embeddings=g("max_position_embeddings", 262144),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class ZeroCenteredRMSNorm(nn.Module):
    """RMSNorm with (1 + weight) scaling, weight zero-init; multiply done in
    float32 before the cast (Qwen3Next convention)."""

    def __init__(self, dim: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.zeros(dim))
        self.eps = eps

    def forward(self, x):
        xf = x.float()
        xf = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + self.eps)
        return (xf * (1.0 + self.weight.float())).type_as(x)


class GatedHeadNorm(nn.Module):
    """Per-head gated RMSNorm (plain weight): norm, weight in input dtype,
    then * silu(gate) in float32."""

    def __init__(self, dim: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, y, z):
        dt = y.dtype
        yf = y.float()
        yf = yf * torch.rsqrt(yf.pow(2).mean(-1, keepdim=True) + self.eps)
        y = self.weight * yf.to(dt)
        return (y * F.silu(z.float())).to(dt)


def _l2norm(x, eps: float = 1e-6):
    return x * torch.rsqrt((x * x).sum(-1, keepdim=True) + eps)


def gated_delta_rule_chunked(q, k, v, g, beta, chunk_size: int = 64,
                             initial_state=None, return_final_state=False):
    """Chunked gated delta rule (float32). q/k/v [B,S,Hv,D*]; g/beta [B,S,Hv].

    Math identical to the reference's torch_chunk_gated_delta_rule
    (modeling_qwen3_next.py:375): l2norm(q,k), beta-weighted rank-1 delta
    updates with per-head log decay g, processed in chunks. The per-row
    forward-substitution loop building T = (I - tril(Kb K^T decay, -1))^{-1}
    is replaced by one batched triangular solve.

    ``initial_state`` [B,H,Dk,Dv] / ``return_final_state`` support the
    linear-attention context-parallel chunk relay (parallel/cp_linear.py).
    """
    q = _l2norm(q.transpose(1, 2).float())
    k = _l2norm(k.transpose(1, 2).float())
    v = v.transpose(1, 2).float()
    beta = beta.transpose(1, 2).float()
    g = g.transpose(1, 2).float()
    b, h, s, dk = k.shape
    dv = v.shape[-1]
    pad = (chunk_size - s % chunk_size) % chunk_size
    if pad:
        q = F.pad(q, (0, 0, 0, pad))
        k = F.pad(k, (0, 0, 0, pad))
        v = F.pad(v, (0, 0, 0, pad))
        beta = F.pad(beta, (0, pad))
        g = F.pad(g, (0, pad))
    n = (s + pad) // chunk_size
    c = chunk_size
    q = q.view(b, h, n, c, dk) * (dk ** -0.5)
    k = k.view(b, h, n, c, dk)
    v = v.view(b, h, n, c, dv)
    beta = beta.view(b, h, n, c)
    g = g.view(b, h, n, c).cumsum(dim=-1)

    v_beta = v * beta.unsqueeze(-1)
    k_beta = k * beta.unsqueeze(-1)
    decay = (g.unsqueeze(-1) - g.unsqueeze(-2)).tril().exp().tril()
    a0 = -((k_beta @ k.transpose(-1, -2)) * decay).tril(-1)
    eye = torch.eye(c, dtype=a0.dtype, device=a0.device).expand_as(a0)
    T = torch.linalg.solve_triangular(eye - a0, eye.contiguous(),
                                      upper=False, unitriangular=True)
    v_in = T @ v_beta
    k_cumdecay = T @ (k_beta * g.exp().unsqueeze(-1))

    state = (q.new_zeros(b, h, dk, dv) if initial_state is None
             else initial_state.to(q.dtype))
    out = torch.empty_like(v_in)
    attn_mask = torch.ones(c, c, dtype=torch.bool, device=q.device).triu(1)
    for i in range(n):
        q_i, k_i, v_i = q[:, :, i], k[:, :, i], v_in[:, :, i]
        attn = (q_i @ k_i.transpose(-1, -2) * decay[:, :, i]).masked_fill(attn_mask, 0)
        v_prime = k_cumdecay[:, :, i] @ state
        v_new = v_i - v_prime
        inter = (q_i * g[:, :, i, :, None].exp()) @ state
        out[:, :, i] = inter + attn @ v_new
        state = (state * g[:, :, i, -1, None, None].exp() +
                 (k_i * (g[:, :, i, -1, None] - g[:, :, i]).exp()[..., None]
                  ).transpose(-1, -2) @ v_new)
    out = out.reshape(b, h, -1, dv)[:, :, :s].transpose(1, 2)
    if return_final_state:
        return out, state
    return out


class GatedDeltaNet(nn.Module):
    """Linear-attention mixer (reference modeling_qwen3_next.py:512)."""

    def __init__(self, cfg: Qwen3NextConfig):
        super().__init__()
        self.num_v_heads = cfg.linear_num_value_heads
        self.num_k_heads = cfg.linear_num_key_heads
        self.head_k_dim = cfg.linear_key_head_dim
        self.head_v_dim = cfg.linear_value_head_dim
        self.key_dim = self.head_k_dim * self.num_k_heads
        self.value_dim = self.head_v_dim * self.num_v_heads
        self.conv_dim = self.key_dim * 2 + self.value_dim
        self.conv_kernel = cfg.linear_conv_kernel_dim
        self.in_proj_qkvz = nn.Linear(cfg.hidden_size,
                                      self.key_dim * 2 + self.value_dim * 2,
                                      bias=False)
        self.in_proj_ba = nn.Linear(cfg.hidden_size, self.num_v_heads * 2, bias=False)
        self.conv1d = nn.Conv1d(self.conv_dim, self.conv_dim, self.conv_kernel,
                                groups=self.conv_dim,
                                padding=self.conv_kernel - 1, bias=False)
        self.dt_bias = nn.Parameter(torch.ones(self.num_v_heads))
        self.A_log = nn.Parameter(torch.zeros(self.num_v_heads))
        self.norm = GatedHeadNorm(self.head_v_dim, eps=cfg.rms_norm_eps)
        self.out_proj = nn.Linear(self.value_dim, cfg.hidden_size, bias=False)

    def _split_qkvzba(self, qkvz, ba):
        """Per-k-head interleaved layout (reference :557)."""
        B, S, _ = qkvz.shape
        r = self.num_v_heads // self.num_k_heads
        dk, dv = self.head_k_dim, self.head_v_dim
        qkvz = qkvz.view(B, S, self.num_k_heads, 2 * dk + 2 * r * dv)
        q, k, v, z = torch.split(qkvz, [dk, dk, r * dv, r * dv], dim=3)
        ba = ba.view(B, S, self.num_k_heads, 2 * r)
        bb, aa = torch.split(ba, [r, r], dim=3)
        v = v.reshape(B, S, -1, dv)
        z = z.reshape(B, S, -1, dv)
        return q, k, v, z, bb.reshape(B, S, -1), aa.reshape(B, S, -1)

    def forward(self, h):
        B, S, _ = h.shape
        q, k, v, z, b, a = self._split_qkvzba(self.in_proj_qkvz(h), self.in_proj_ba(h))
        qkv = torch.cat([q.reshape(B, S, -1), k.reshape(B, S, -1),
                         v.reshape(B, S, -1)], dim=-1)
        qkv = F.silu(self.conv1d(qkv.transpose(1, 2))[..., :S].transpose(1, 2))
        q, k, v = torch.split(qkv, [self.key_dim, self.key_dim, self.value_dim], dim=-1)
        q = q.reshape(B, S, -1, self.head_k_dim)
        k = k.reshape(B, S, -1, self.head_k_dim)
        v = v.reshape(B, S, -1, self.head_v_dim)
        beta = b.sigmoid()
        g = -self.A_log.float().exp() * F.softplus(a.float() + self.dt_bias)
        r = self.num_v_heads // self.num_k_heads
        if r > 1:
            q = q.repeat_interleave(r, dim=2)
            k = k.repeat_interleave(r, dim=2)
        o = gated_delta_rule_chunked(q, k, v, g, beta).to(h.dtype)
        o = self.norm(o.reshape(-1, self.head_v_dim), z.reshape(-1, self.head_v_dim))
        return self.out_proj(o.view(B, S, -1))


class Qwen3NextAttention(nn.Module):
    """GQA with sigmoid output gate carved out of q_proj, zero-centered
    per-head q/k norms, partial rotary (reference :236)."""

    def __init__(self, cfg: Qwen3NextConfig, backend: BackendConfig):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.head_dim = D
        self.rot = int(D * cfg.partial_rotary_factor)
        bq = cfg.attention_bias
        self.q_proj = nn.Linear(cfg.hidden_size, H * D * 2, bias=bq)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=bq)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=bq)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=bq)
        self.q_norm = ZeroCenteredRMSNorm(D, cfg.rms_norm_eps)
        self.k_norm = ZeroCenteredRMSNorm(D, cfg.rms_norm_eps)
        self.backend = backend

    def forward(self, h, cos, sin):
        B, S, _ = h.shape
        D = self.head_dim
        qg = self.q_proj(h).view(B, S, -1, 2 * D)
        q, gate = qg.chunk(2, dim=-1)
        gate = gate.reshape(B, S, -1)
        q = self.q_norm(q)
        k = self.k_norm(self.k_proj(h).view(B, S, -1, D))
        v = self.v_proj(h).view(B, S, -1, D)
        r = self.rot
        if r < D:
            qr, kr = apply_rope(q[..., :r].contiguous(), k[..., :r].contiguous(),
                                cos, sin, backend="torch")
            q = torch.cat([qr, q[..., r:]], dim=-1)
            k = torch.cat([kr, k[..., r:]], dim=-1)
        else:
            q, k = apply_rope(q, k, cos, sin, backend=self.backend.rope)
        o = flash_attention(q, k, v, causal=True, backend=self.backend.attn)
        o = o.reshape(B, S, -1) * torch.sigmoid(gate)
        return self.o_proj(o)


class Qwen3NextMLP(nn.Module):
    def __init__(self, hidden: int, inter: int):
        super().__init__()
        self.gate_proj = nn.Linear(hidden, inter, bias=False)
        self.up_proj = nn.Linear(hidden, inter, bias=False)
        self.down_proj = nn.Linear(inter, hidden, bias=False)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


class Qwen3NextDecoderLayer(nn.Module):
    def __init__(self, cfg: Qwen3NextConfig, backend: BackendConfig, layer_idx: int):
        super().__init__()
        types = cfg.layer_types or [
            "full_attention" if (i + 1) % 4 == 0 else "linear_attention"
            for i in range(cfg.num_hidden_layers)]
        self.is_attn = types[layer_idx] == "full_attention"
        if self.is_attn:
            self.self_attn = Qwen3NextAttention(cfg, backend)
        else:
            self.linear_attn = GatedDeltaNet(cfg)
        sparse = (layer_idx not in (cfg.mlp_only_layers or []) and
                  cfg.num_experts > 0 and
                  (layer_idx + 1) % cfg.decoder_sparse_step == 0)
        if sparse:
            self.mlp = MoE(cfg.hidden_size, MoEConfig(
                n_routed_experts=cfg.num_experts,
                n_activated_experts=cfg.num_experts_per_tok,
                moe_intermediate_size=cfg.moe_intermediate_size,
                n_shared_experts=1,
                shared_expert_intermediate_size=cfg.shared_expert_intermediate_size,
                shared_expert_gate=True,
                norm_topk_prob=cfg.norm_topk_prob,
                aux_loss_coeff=cfg.router_aux_loss_coef))
        else:
            self.mlp = Qwen3NextMLP(cfg.hidden_size, cfg.intermediate_size)
        self.input_layernorm = ZeroCenteredRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.post_attention_layernorm = ZeroCenteredRMSNorm(cfg.hidden_size,
                                                            cfg.rms_norm_eps)

    def forward(self, x, cos, sin):
        h = self.input_layernorm(x)
        h = self.self_attn(h, cos, sin) if self.is_attn else self.linear_attn(h)
        x = x + h
        return x + self.mlp(self.post_attention_layernorm(x))


class Qwen3NextModel(nn.Module):
    def __init__(self, cfg: Qwen3NextConfig, backend: BackendConfig):
        super().__init__()
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            Qwen3NextDecoderLayer(cfg, backend, i)
            for i in range(cfg.num_hidden_layers))
        self.norm = ZeroCenteredRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        rot = int(cfg.head_dim * cfg.partial_rotary_factor)
        cos, sin = build_rope_cache(rot, min(cfg.max_position_embeddings, 32768),
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


class Qwen3NextStateDictAdapter:
    """HF fused gate_up_proj [E,2I,H] <-> split stacked [E,I,H]; HF
    mlp.shared_expert.* <-> mlp.shared_experts.*."""

    def from_hf(self, sd: dict) -> dict:
        out = {}
        for k, v in sd.items():
            if k.endswith("mlp.experts.gate_up_proj"):
                gate, up = v.chunk(2, dim=1)
                out[k.replace("gate_up_proj", "gate_proj")] = gate.contiguous()
                out[k.replace("gate_up_proj", "up_proj")] = up.contiguous()
            elif ".mlp.shared_expert." in k:
                out[k.replace(".mlp.shared_expert.", ".mlp.shared_experts.")] = v
            else:
                out[k] = v
        return out

    def to_hf(self, sd: dict) -> dict:
        out = {}
        for k, v in sd.items():
            if k.endswith("mlp.experts.gate_proj"):
                up = sd[k.replace("gate_proj", "up_proj")]
                out[k.replace("gate_proj", "gate_up_proj")] = torch.cat([v, up], dim=1)
            elif k.endswith("mlp.experts.up_proj"):
                continue
            elif ".mlp.shared_experts." in k:
                out[k.replace(".mlp.shared_experts.", ".mlp.shared_expert.")] = v
            else:
                out[k] = v
        return out


class Qwen3NextForCausalLM(nn.Module):
    hf_architectures = ("Qwen3NextForCausalLM",)
    config_class = Qwen3NextConfig
    state_dict_adapter = Qwen3NextStateDictAdapter

    @staticmethod
    def config_from_hf(hf_cfg) -> Qwen3NextConfig:
        return Qwen3NextConfig.from_hf_config(hf_cfg)

    def __init__(self, config: Qwen3NextConfig | dict, backend=None):
        super().__init__()
        cfg = (config if isinstance(config, Qwen3NextConfig)
               else Qwen3NextConfig(**dict(config)))
        self.config = cfg
        bk = BackendConfig.resolve(backend,
                                   "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.head_dim)
        self.model = Qwen3NextModel(cfg, bk)
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
            rot = int(cfg.head_dim * cfg.partial_rotary_factor)
            cos, sin = build_rope_cache(rot, min(cfg.max_position_embeddings, 32768),
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
            elif isinstance(m, ZeroCenteredRMSNorm):
                nn.init.zeros_(m.weight)
            elif isinstance(m, GatedHeadNorm):
                nn.init.ones_(m.weight)
            elif isinstance(m, GatedDeltaNet):
                nn.init.ones_(m.dt_bias)
                m.A_log.copy_(torch.empty_like(m.A_log).uniform_(0, 16).log())
            elif isinstance(m, MoE):
                nn.init.normal_(m.gate.weight, std=std)
                nn.init.normal_(m.experts.gate_proj, std=std)
                nn.init.normal_(m.experts.up_proj, std=std)
                nn.init.normal_(m.experts.down_proj, std=std)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
