"""Jamba (AI21 hybrid Mamba-1 + attention + MoE) causal LM, MI355X-native.

Reference behavior: the public Jamba architecture (HF
transformers.models.jamba) — period/offset-scheduled attention (NoPE GQA)
vs Mamba-1 mixers (selective scan with dt/B/C RMSNorms — Jamba's key
difference from vanilla Mamba), and period/offset-scheduled MoE
(softmax-then-topk, NO renorm) vs dense SwiGLU FFNs.

The Mamba-1 selective scan here is a chunked log-space segsum formulation
(the diagonal recurrence h_t = exp(dt_t A) h_{t-1} + dt_t B_t x_t solved
per chunk with pairwise decays exp(cum_t - cum_s) <= 1 — numerically safe,
GEMM/elementwise-shaped for the GPU instead of the reference's per-step
python recurrence). Pairwise chunk memory is O(c^2 D N); chunk=16 default.
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
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.swiglu import swiglu


def mamba1_chunked_scan(x, dt, A, B, C, chunk: int = 16):
    """x/dt [B,S,D] (dt post-softplus), A [D,N], B/C [B,S,N] -> y [B,S,D]."""
    Bb, S, D = x.shape
    N = A.shape[1]
    pad = (chunk - S % chunk) % chunk
    if pad:
        x = F.pad(x, (0, 0, 0, pad))
        dt = F.pad(dt, (0, 0, 0, pad))
        B = F.pad(B, (0, 0, 0, pad))
        C = F.pad(C, (0, 0, 0, pad))
    n = (S + pad) // chunk
    la = (dt[..., None].float() * A).view(Bb, n, chunk, D, N)
    b = ((dt * x)[..., None].float() * B[:, :, None, :].float()
         ).view(Bb, n, chunk, D, N)
    Cc = C.float().view(Bb, n, chunk, N)
    cum = la.cumsum(2)
    # pairwise decay within the chunk: exp(cum_t - cum_s), t >= s (<= 1);
    # mask BEFORE exp — the t<s half is exp(positive) and would inf*0=NaN
    mask = torch.ones(chunk, chunk, dtype=torch.bool, device=x.device).tril()
    dec = (cum.unsqueeze(3) - cum.unsqueeze(2)).masked_fill(
        ~mask.view(1, 1, chunk, chunk, 1, 1), float("-inf")).exp()
    h = torch.einsum("bctsdn,bcsdn->bctdn", dec, b)
    state = x.new_zeros(Bb, D, N, dtype=torch.float32)
    ys = []
    for i in range(n):
        hi = h[:, i] + cum[:, i].exp() * state[:, None]
        ys.append(torch.einsum("btdn,btn->btd", hi, Cc[:, i]))
        state = hi[:, -1]
    return torch.cat(ys, dim=1)[:, :S]


class JambaMambaMixer(nn.Module):
    """Mamba-1 selective mixer with Jamba's dt/B/C RMSNorms."""

    def __init__(self, hidden_size: int, intermediate: int, state_size: int,
                 dt_rank: int, conv_kernel: int = 4, eps: float = 1e-6,
                 use_bias: bool = False, use_conv_bias: bool = True,
                 dt_bc_norms: str = "weighted"):
        """dt_bc_norms: "weighted" (Jamba), "weightless" (FalconMamba),
        "none" (vanilla Mamba)."""
        super().__init__()
        self.inter = intermediate
        self.state_size = state_size
        self.dt_rank = dt_rank
        self.dt_bc_norms = dt_bc_norms
        self.eps = eps
        self.in_proj = nn.Linear(hidden_size, 2 * intermediate, bias=use_bias)
        self.conv1d = nn.Conv1d(intermediate, intermediate, conv_kernel,
                                groups=intermediate, padding=conv_kernel - 1,
                                bias=use_conv_bias)
        self.x_proj = nn.Linear(intermediate, dt_rank + 2 * state_size, bias=False)
        self.dt_proj = nn.Linear(dt_rank, intermediate, bias=True)
        if dt_bc_norms == "weighted":
            self.dt_layernorm = RMSNorm(dt_rank, eps, "torch")
            self.b_layernorm = RMSNorm(state_size, eps, "torch")
            self.c_layernorm = RMSNorm(state_size, eps, "torch")
        self.A_log = nn.Parameter(torch.zeros(intermediate, state_size))
        self.D = nn.Parameter(torch.ones(intermediate))
        self.out_proj = nn.Linear(intermediate, hidden_size, bias=use_bias)

    @staticmethod
    def _rms(t: torch.Tensor, eps: float) -> torch.Tensor:
        tf = t.float()
        return (tf * torch.rsqrt(tf.pow(2).mean(-1, keepdim=True) + eps)).to(t.dtype)

    def forward(self, h: torch.Tensor) -> torch.Tensor:
        S = h.shape[1]
        x, z = self.in_proj(h).chunk(2, dim=-1)
        x = F.silu(self.conv1d(x.transpose(1, 2))[..., :S].transpose(1, 2))
        dt_r, B, C = self.x_proj(x).split(
            [self.dt_rank, self.state_size, self.state_size], dim=-1)
        if self.dt_bc_norms == "weighted":
            dt_r = self.dt_layernorm(dt_r)
            B = self.b_layernorm(B)
            C = self.c_layernorm(C)
        elif self.dt_bc_norms == "weightless":
            dt_r = self._rms(dt_r, self.eps)
            B = self._rms(B, self.eps)
            C = self._rms(C, self.eps)
        dt = F.softplus(F.linear(dt_r, self.dt_proj.weight).float()
                        + self.dt_proj.bias.float())
        A = -torch.exp(self.A_log.float())
        y = mamba1_chunked_scan(x, dt, A, B, C)
        y = y + self.D.float() * x.float()
        y = (y * F.silu(z.float())).to(h.dtype)
        return self.out_proj(y)


class JambaAttention(nn.Module):
    """Plain NoPE GQA (Jamba attention layers carry no positional encoding)."""

    def __init__(self, cfg: "JambaConfig", backend: BackendConfig):
        super().__init__()
        H, Hk = cfg.num_attention_heads, cfg.num_key_value_heads
        D = cfg.hidden_size // H
        self.head_dim = D
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=False)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        self.backend = backend

    def forward(self, h):
        B, S, _ = h.shape
        D = self.head_dim
        q = self.q_proj(h).view(B, S, -1, D)
        k = self.k_proj(h).view(B, S, -1, D)
        v = self.v_proj(h).view(B, S, -1, D)
        o = flash_attention(q, k, v, causal=True, backend=self.backend.attn)
        return self.o_proj(o.reshape(B, S, -1))


class JambaMLP(nn.Module):
    def __init__(self, cfg: "JambaConfig"):
        super().__init__()
        self.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        self.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


@dataclass
class JambaConfig:
    vocab_size: int = 65536
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    mamba_d_state: int = 16
    mamba_d_conv: int = 4
    mamba_expand: int = 2
    mamba_dt_rank: int | str = 256
    mamba_conv_bias: bool = True
    mamba_proj_bias: bool = False
    num_experts: int = 16
    num_experts_per_tok: int = 2
    expert_layer_period: int = 2
    expert_layer_offset: int = 1
    attn_layer_period: int = 8
    attn_layer_offset: int = 4
    router_aux_loss_coef: float = 0.001
    rms_norm_eps: float = 1e-6
    max_position_embeddings: int = 262144
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    def __post_init__(self):
        if self.mamba_dt_rank == "auto":
            self.mamba_dt_rank = -(-self.hidden_size // 16)

    @classmethod
    def from_hf_config(cls, hf: Any) -> "JambaConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        return cls(
            vocab_size=g("vocab_size", 65536),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 14336),
            num_hidden_layers=g("num_hidden_layers", 32),
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads", 8),
            mamba_d_state=g("mamba_d_state", 16),
            mamba_d_conv=g("mamba_d_conv", 4),
            mamba_expand=g("mamba_expand", 2),
            mamba_dt_rank=g("mamba_dt_rank", "auto"),
            mamba_conv_bias=g("mamba_conv_bias", True),
            mamba_proj_bias=g("mamba_proj_bias", False),
            num_experts=g("num_experts", 16),
            num_experts_per_tok=g("num_experts_per_tok", 2),
            expert_layer_period=g("expert_layer_period", 2),
            expert_layer_offset=g("expert_layer_offset", 1),
            attn_layer_period=g("attn_layer_period", 8),
            attn_layer_offset=g("attn_layer_offset", 4),
            router_aux_loss_coef=g("router_aux_loss_coef", 0.001),
            rms_norm_eps=g("rms_norm_eps", 1e-6),
            max_position_embeddings=g("max_position_embeddings", 262144),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class JambaDecoderLayer(nn.Module):
    def __init__(self, cfg: JambaConfig, backend: BackendConfig, layer_idx: int):
        super().__init__()
        self.is_attn = (layer_idx % cfg.attn_layer_period == cfg.attn_layer_offset)
        is_moe = (cfg.num_experts > 1
                  and layer_idx % cfg.expert_layer_period == cfg.expert_layer_offset)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        self.pre_ff_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        if self.is_attn:
            self.self_attn = JambaAttention(cfg, backend)
        else:
            self.mamba = JambaMambaMixer(
                cfg.hidden_size, cfg.mamba_expand * cfg.hidden_size,
                cfg.mamba_d_state, cfg.mamba_dt_rank, cfg.mamba_d_conv,
                cfg.rms_norm_eps, use_bias=cfg.mamba_proj_bias,
                use_conv_bias=cfg.mamba_conv_bias)
        if is_moe:
            self.feed_forward = MoE(cfg.hidden_size, MoEConfig(
                n_routed_experts=cfg.num_experts,
                n_activated_experts=cfg.num_experts_per_tok,
                moe_intermediate_size=cfg.intermediate_size,
                norm_topk_prob=False,
                aux_loss_coeff=cfg.router_aux_loss_coef))
        else:
            self.feed_forward = JambaMLP(cfg)

    def forward(self, x):
        h = self.input_layernorm(x)
        h = self.self_attn(h) if self.is_attn else self.mamba(h)
        x = x + h
        return x + self.feed_forward(self.pre_ff_layernorm(x))


class JambaModel(nn.Module):
    def __init__(self, cfg: JambaConfig, backend: BackendConfig):
        super().__init__()
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            JambaDecoderLayer(cfg, backend, i) for i in range(cfg.num_hidden_layers))
        self.final_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)

    def forward(self, ids):
        x = self.embed_tokens(ids)
        for layer in self.layers:
            x = layer(x)
        return self.final_layernorm(x)


class JambaStateDictAdapter:
    """HF feed_forward.router -> feed_forward.gate; fused gate_up split."""

    def from_hf(self, sd: dict) -> dict:
        out = {}
        for k, v in sd.items():
            if k.endswith("feed_forward.router.weight"):
                out[k.replace(".router.weight", ".gate.weight")] = v
            elif k.endswith("feed_forward.experts.gate_up_proj"):
                gate, up = v.chunk(2, dim=1)
                out[k.replace("gate_up_proj", "gate_proj")] = gate.contiguous()
                out[k.replace("gate_up_proj", "up_proj")] = up.contiguous()
            else:
                out[k] = v
        return out

    def to_hf(self, sd: dict) -> dict:
        out = {}
        for k, v in sd.items():
            if k.endswith("feed_forward.gate.weight"):
                out[k.replace(".gate.weight", ".router.weight")] = v
            elif k.endswith("feed_forward.experts.gate_proj"):
                up = sd[k.replace("gate_proj", "up_proj")]
                out[k.replace("gate_proj", "gate_up_proj")] = torch.cat([v, up], dim=1)
            elif k.endswith("feed_forward.experts.up_proj"):
                continue
            else:
                out[k] = v
        return out


class JambaForCausalLM(nn.Module):
    hf_architectures = ("JambaForCausalLM",)
    config_class = JambaConfig
    state_dict_adapter = JambaStateDictAdapter

    @staticmethod
    def config_from_hf(hf_cfg) -> JambaConfig:
        return JambaConfig.from_hf_config(hf_cfg)

    def __init__(self, config: JambaConfig | dict, backend=None):
        super().__init__()
        cfg = config if isinstance(config, JambaConfig) else JambaConfig(**dict(config))
        self.config = cfg
        bk = BackendConfig.resolve(
            backend, "cuda" if torch.cuda.is_available() else "cpu",
            head_dim=cfg.hidden_size // cfg.num_attention_heads)
        self.model = JambaModel(cfg, bk)
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
            return F.cross_entropy(
                logits.reshape(-1, logits.shape[-1]).float(), labels.reshape(-1),
                ignore_index=-100, reduction="sum")
        return logits

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        cfg = self.config
        if device is not None:
            self.to_empty(device=device)
        std = cfg.initializer_range
        for m in self.modules():
            if isinstance(m, (nn.Linear, nn.Conv1d)):
                nn.init.normal_(m.weight, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif type(m).__name__ == "RMSNorm":
                nn.init.ones_(m.weight)
            elif isinstance(m, JambaMambaMixer):
                # S4D real init: A_log[d, n] = log(n + 1)
                A = torch.arange(1, m.state_size + 1, dtype=torch.float32,
                                 device=m.A_log.device).expand(m.inter, -1)
                m.A_log.copy_(A.log())
                nn.init.ones_(m.D)
            elif isinstance(m, MoE):
                nn.init.normal_(m.gate.weight, std=std)
                nn.init.normal_(m.experts.gate_proj, std=std)
                nn.init.normal_(m.experts.up_proj, std=std)
                nn.init.normal_(m.experts.down_proj, std=std)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
