"""RecurrentGemma (Griffin: RG-LRU recurrent blocks + windowed attention).

Reference behavior: the public RecurrentGemma architecture (HF
transformers.models.recurrent_gemma) — alternating temporal blocks
(recurrent: conv1d + Real-Gated LRU with block-diagonal per-head gates;
attention: partial-rotary sliding-window GQA), biased GeGLU MLPs at
intermediate//2 width, bf16-rounded sqrt(H) embedding normalizer, tanh
logits soft-cap. The RG-LRU scan is the same chunked log-space segsum used
by the Mamba mixers (pairwise decays exp(cum_t - cum_s) <= 1), with the
Griffin sqrt(1 - a^2) input normalization and position-0 reset.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.ops.attention import flash_attention
from automodel_amd.models.qwen3_next.model import ZeroCenteredRMSNorm
from automodel_amd.ops.rope import apply_rope, build_rope_cache


def rg_lru_scan(x: torch.Tensor, log_a: torch.Tensor, chunk: int = 64):
    """Diagonal linear recurrence h_t = a_t h_{t-1} + x_t with a = exp(log_a)
    (log_a <= 0). x/log_a [B, S, D] -> [B, S, D], float32 chunked segsum."""
    B, S, D = x.shape
    pad = (chunk - S % chunk) % chunk
    if pad:
        x = F.pad(x, (0, 0, 0, pad))
        log_a = F.pad(log_a, (0, 0, 0, pad))
    n = (S + pad) // chunk
    xf = x.float().view(B, n, chunk, D)
    la = log_a.float().view(B, n, chunk, D)
    cum = la.cumsum(2)
    mask = torch.ones(chunk, chunk, dtype=torch.bool, device=x.device).tril()
    dec = (cum.unsqueeze(3) - cum.unsqueeze(2)).masked_fill(
        ~mask.view(1, 1, chunk, chunk, 1), float("-inf")).exp()
    h = torch.einsum("bcts,bcsd->bctd", dec.squeeze(-1), xf) \
        if D == 1 else torch.einsum("bctsd,bcsd->bctd", dec, xf)
    state = x.new_zeros(B, D, dtype=torch.float32)
    outs = []
    for i in range(n):
        hi = h[:, i] + cum[:, i].exp() * state[:, None]
        outs.append(hi)
        state = hi[:, -1]
    return torch.cat(outs, dim=1)[:, :S]


@dataclass
class RecurrentGemmaConfig:
    vocab_size: int = 256000
    hidden_size: int = 2560
    intermediate_size: int = 15360       # MLP width is intermediate // 2
    num_hidden_layers: int = 26
    num_attention_heads: int = 10
    num_key_value_heads: int = 1
    head_dim: int = 256
    lru_width: int | None = 2560
    conv1d_width: int = 4
    attention_window_size: int = 2048
    block_types: tuple = ("recurrent", "recurrent", "attention")
    partial_rotary_factor: float = 1.0
    rope_theta: float = 10000.0
    logits_soft_cap: float = 30.0
    rms_norm_eps: float = 1e-6
    hidden_activation: str = "gelu_pytorch_tanh"
    max_position_embeddings: int = 8192
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    @property
    def layers_block_type(self) -> list[str]:
        bt = list(self.block_types)
        return [bt[i % len(bt)] for i in range(self.num_hidden_layers)]

    @classmethod
    def from_hf_config(cls, hf: Any) -> "RecurrentGemmaConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        rp = g("rope_parameters") or {}
        return cls(
            vocab_size=g("vocab_size", 256000),
            hidden_size=g("hidden_size", 2560),
            intermediate_size=g("intermediate_size", 15360),
            num_hidden_layers=g("num_hidden_layers", 26),
            num_attention_heads=g("num_attention_heads", 10),
            num_key_value_heads=g("num_key_value_heads", 1),
            head_dim=g("head_dim", 256),
            lru_width=g("lru_width") or g("hidden_size", 2560),
            conv1d_width=g("conv1d_width", 4),
            attention_window_size=g("attention_window_size", 2048),
            block_types=tuple(g("block_types") or ("recurrent", "recurrent", "attention")),
            # the reference's rope init takes dim straight from head_dim and
            # IGNORES partial_rotary_factor entirely (its local
            # compute_default_rope_parameters, modeling_recurrent_gemma.py:98)
            # — so rotary is always full-width
            partial_rotary_factor=1.0,
            rope_theta=rp.get("rope_theta", g("rope_theta", 10000.0)),
            logits_soft_cap=g("logits_soft_cap", 30.0),
            rms_norm_eps=g("rms_norm_eps", 1e-6),
            hidden_activation=g("hidden_activation", "gelu_pytorch_tanh"),
            max_position_embeddings=g("max_position_embeddings", 8192),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


def _act(name: str):
    return (lambda t: F.gelu(t, approximate="tanh")) if name == "gelu_pytorch_tanh" \
        else getattr(F, name)


class RgLru(nn.Module):
    """Real-Gated LRU with block-diagonal per-head gate projections."""

    def __init__(self, cfg: RecurrentGemmaConfig):
        super().__init__()
        H = cfg.num_attention_heads
        bw = cfg.lru_width // H
        self.n_heads, self.bw = H, bw
        self.recurrent_param = nn.Parameter(torch.empty(cfg.lru_width))
        self.input_gate_weight = nn.Parameter(torch.empty(H, bw, bw))
        self.input_gate_bias = nn.Parameter(torch.empty(H, bw))
        self.recurrent_gate_weight = nn.Parameter(torch.empty(H, bw, bw))
        self.recurrent_gate_bias = nn.Parameter(torch.empty(H, bw))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, W = x.shape
        a = x.reshape(B * S, self.n_heads, self.bw).permute(1, 0, 2)
        ig = torch.sigmoid(torch.baddbmm(self.input_gate_bias[:, None], a,
                                         self.input_gate_weight)
                           .transpose(0, 1).reshape(B, S, W))
        rg = torch.sigmoid(torch.baddbmm(self.recurrent_gate_bias[:, None], a,
                                         self.recurrent_gate_weight)
                           .transpose(0, 1).reshape(B, S, W))
        log_a = -8.0 * rg * F.softplus(self.recurrent_param)
        a_sq = torch.exp(2 * log_a)
        mult = torch.sqrt((1 - a_sq).clamp_min(1e-6))
        # position 0 resets the state: full passthrough, no decay history
        reset = torch.zeros(B, S, 1, dtype=torch.bool, device=x.device)
        reset[:, 0] = True
        mult = torch.where(reset, torch.ones_like(mult), mult)
        xn = x * ig * mult.to(x.dtype)
        return rg_lru_scan(xn, log_a).to(x.dtype)


class RecurrentBlock(nn.Module):
    def __init__(self, cfg: RecurrentGemmaConfig):
        super().__init__()
        W = cfg.lru_width
        self.linear_y = nn.Linear(cfg.hidden_size, W)
        self.linear_x = nn.Linear(cfg.hidden_size, W)
        self.linear_out = nn.Linear(W, cfg.hidden_size)
        self.conv_1d = nn.Conv1d(W, W, cfg.conv1d_width, groups=W,
                                 padding=cfg.conv1d_width - 1)
        self.rg_lru = RgLru(cfg)
        self.act = _act(cfg.hidden_activation)

    def forward(self, h, cos, sin):
        S = h.shape[1]
        y = self.act(self.linear_y(h))
        x = self.conv_1d(self.linear_x(h).transpose(1, 2))[..., :S].transpose(1, 2)
        return self.linear_out(self.rg_lru(x) * y)


class WindowedAttention(nn.Module):
    def __init__(self, cfg: RecurrentGemmaConfig, backend: BackendConfig):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.head_dim = D
        self.rot = int(D * cfg.partial_rotary_factor)
        self.window = cfg.attention_window_size
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=False)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=False)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=True)
        self.backend = backend

    def forward(self, h, cos, sin):
        B, S, _ = h.shape
        D = self.head_dim
        q = self.q_proj(h).view(B, S, -1, D)
        k = self.k_proj(h).view(B, S, -1, D)
        v = self.v_proj(h).view(B, S, -1, D)
        r = self.rot
        qr, kr = apply_rope(q[..., :r].contiguous(), k[..., :r].contiguous(),
                            cos, sin, backend="torch")
        q = torch.cat([qr, q[..., r:]], dim=-1)
        k = torch.cat([kr, k[..., r:]], dim=-1)
        if S <= self.window:
            o = flash_attention(q, k, v, causal=True, backend=self.backend.attn)
        else:
            # banded causal mask: key j visible to query i iff i-window < j <= i
            from automodel_amd.ops.attention import sdpa_masked

            i = torch.arange(S, device=h.device)
            allowed = (i[None, :] <= i[:, None]) & (i[None, :] > i[:, None] - self.window)
            mask = torch.where(allowed, 0.0, float("-inf")) \
                .to(q.dtype).view(1, 1, S, S)
            o = sdpa_masked(q, k, v, mask)
        return self.o_proj(o.reshape(B, S, -1))


class RecurrentGemmaMlp(nn.Module):
    def __init__(self, cfg: RecurrentGemmaConfig):
        super().__init__()
        inter = cfg.intermediate_size // 2
        self.gate_proj = nn.Linear(cfg.hidden_size, inter, bias=True)
        self.up_proj = nn.Linear(cfg.hidden_size, inter, bias=True)
        self.down_proj = nn.Linear(inter, cfg.hidden_size, bias=True)
        self.act = _act(cfg.hidden_activation)

    def forward(self, x):
        return self.down_proj(self.act(self.gate_proj(x)) * self.up_proj(x))


class RecurrentGemmaLayer(nn.Module):
    def __init__(self, cfg: RecurrentGemmaConfig, backend: BackendConfig, i: int):
        super().__init__()
        self.temporal_pre_norm = ZeroCenteredRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        kind = cfg.layers_block_type[i]
        self.temporal_block = (WindowedAttention(cfg, backend) if kind == "attention"
                               else RecurrentBlock(cfg))
        self.channel_pre_norm = ZeroCenteredRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.mlp_block = RecurrentGemmaMlp(cfg)

    def forward(self, x, cos, sin):
        h = self.temporal_block(self.temporal_pre_norm(x), cos, sin)
        res = h + x
        return res + self.mlp_block(self.channel_pre_norm(res))


class RecurrentGemmaModel(nn.Module):
    def __init__(self, cfg: RecurrentGemmaConfig, backend: BackendConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            RecurrentGemmaLayer(cfg, backend, i) for i in range(cfg.num_hidden_layers))
        self.final_norm = ZeroCenteredRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        rot = int(cfg.head_dim * cfg.partial_rotary_factor)
        cos, sin = build_rope_cache(rot, min(cfg.max_position_embeddings, 32768),
                                    cfg.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, ids):
        x = self.embed_tokens(ids)
        # bf16-rounded sqrt(H) normalizer (reference keeps it as a bf16 buffer)
        norm = torch.tensor(self.cfg.hidden_size ** 0.5,
                            dtype=torch.bfloat16).to(x.dtype)
        x = x * norm
        S = x.shape[1]
        cos, sin = self.rope_cos[:S].float(), self.rope_sin[:S].float()
        for layer in self.layers:
            x = layer(x, cos, sin)
        return self.final_norm(x)


class RecurrentGemmaForCausalLM(nn.Module):
    hf_architectures = ("RecurrentGemmaForCausalLM",)
    config_class = RecurrentGemmaConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> RecurrentGemmaConfig:
        return RecurrentGemmaConfig.from_hf_config(hf_cfg)

    def __init__(self, config: RecurrentGemmaConfig | dict, backend=None):
        super().__init__()
        cfg = (config if isinstance(config, RecurrentGemmaConfig)
               else RecurrentGemmaConfig(**dict(config)))
        self.config = cfg
        bk = BackendConfig.resolve(backend,
                                   "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.head_dim)
        self.model = RecurrentGemmaModel(cfg, bk)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None,
                return_hidden=False, **_):
        h = self.model(input_ids)
        if return_hidden:
            return h
        logits = self.lm_head(h)
        cap = self.config.logits_soft_cap
        logits = torch.tanh(logits / cap) * cap
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
            rot = int(cfg.head_dim * cfg.partial_rotary_factor)
            cos, sin = build_rope_cache(rot, min(cfg.max_position_embeddings, 32768),
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
            elif isinstance(m, ZeroCenteredRMSNorm):
                nn.init.zeros_(m.weight)
            elif isinstance(m, RgLru):
                nn.init.uniform_(m.recurrent_param, 0.5, 1.5)
                nn.init.normal_(m.input_gate_weight, std=std)
                nn.init.normal_(m.recurrent_gate_weight, std=std)
                nn.init.zeros_(m.input_gate_bias)
                nn.init.zeros_(m.recurrent_gate_bias)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
