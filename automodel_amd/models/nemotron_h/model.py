"""Nemotron-H hybrid Mamba2/attention/MLP causal LM, MI355X-native.

Reference behavior: nemo_automodel/components/models/nemotron_v3 (hybrid
Mamba) and the public NemotronH architecture (HF transformers
models/nemotron_h). Layer pattern comes from ``hybrid_override_pattern``:
'M' = Mamba2 mixer, '*' = GQA attention (no rope), '-' = relu^2 MLP.

The Mamba2 mixer implements the chunked SSD scan in pure torch (the Mamba2
paper's segment-sum formulation) — numerically parity-tested against the HF
model end to end. HIP acceleration of the scan is a follow-up; the
projections, conv and norms already ride hipBLASLt / in-tree kernels.
Key names mirror the HF layout exactly (identity state-dict adapter):
model.embeddings / model.layers.N.{norm,mixer.*} / model.norm_f / lm_head.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rms_norm import RMSNorm


@dataclass
class NemotronHConfig:
    vocab_size: int = 131072
    hidden_size: int = 4096
    intermediate_size: int = 21504
    num_hidden_layers: int = 52
    hybrid_override_pattern: str | None = None
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    head_dim: int = 128
    mamba_num_heads: int = 128
    mamba_head_dim: int = 64
    ssm_state_size: int = 128
    conv_kernel: int = 4
    n_groups: int = 8
    chunk_size: int = 128
    use_conv_bias: bool = True
    use_bias: bool = False
    mlp_bias: bool = False
    attention_bias: bool = False
    layer_norm_epsilon: float = 1e-5
    time_step_limit: tuple = (0.0, float("inf"))
    max_position_embeddings: int = 8192
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    def __post_init__(self):
        if self.hybrid_override_pattern is None:
            self.hybrid_override_pattern = "M" * self.num_hidden_layers

    @property
    def mamba_intermediate(self) -> int:
        return self.mamba_num_heads * self.mamba_head_dim

    @classmethod
    def from_hf_config(cls, hf: Any) -> "NemotronHConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        # the pattern serializes as layers_block_type (linear_attention /
        # full_attention / mlp); hybrid_override_pattern itself is not in
        # to_dict()
        pat = g("hybrid_override_pattern")
        if not pat:
            kinds = {"linear_attention": "M", "mamba": "M",
                     "full_attention": "*", "attention": "*", "mlp": "-"}
            blocks = g("layers_block_type") or []
            pat = "".join(kinds[b] for b in blocks) if blocks else None
        # num_hidden_layers is excluded from NemotronH's to_dict(); fall
        # back to the pattern length
        n_layers = g("num_hidden_layers") or (len(pat) if pat else 52)
        if not pat:
            pat = "M" * n_layers
        return cls(
            vocab_size=g("vocab_size", 131072),
            hidden_size=g("hidden_size", 4096),
            intermediate_size=g("intermediate_size", 21504),
            num_hidden_layers=n_layers,
            hybrid_override_pattern=pat,
            num_attention_heads=g("num_attention_heads", 32),
            num_key_value_heads=g("num_key_value_heads", 8),
            head_dim=g("head_dim", 128),
            mamba_num_heads=g("mamba_num_heads", 128),
            mamba_head_dim=g("mamba_head_dim", 64),
            ssm_state_size=g("ssm_state_size", 128),
            conv_kernel=g("conv_kernel", 4),
            n_groups=g("n_groups", 8),
            chunk_size=g("chunk_size", 128),
            use_conv_bias=g("use_conv_bias", True),
            use_bias=g("use_bias", False),
            mlp_bias=g("mlp_bias", False),
            attention_bias=g("attention_bias", False),
            layer_norm_epsilon=g("layer_norm_epsilon", 1e-5),
            max_position_embeddings=g("max_position_embeddings", 8192),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


def _segsum_decay(a: torch.Tensor) -> torch.Tensor:
    """a [..., L] -> decay [..., L, L]: exp(sum_{k=j+1..i} a_k) for i>=j,
    0 above the diagonal (stable cumsum-difference form)."""
    L = a.shape[-1]
    cs = a.cumsum(-1)
    seg = cs[..., :, None] - cs[..., None, :]          # sum_(j..i]
    mask = torch.tril(torch.ones(L, L, dtype=torch.bool, device=a.device))
    return torch.exp(seg.masked_fill(~mask, float("-inf")))


def mamba2_chunked_scan(x, dt, A, B, C, chunk: int):
    """Chunked SSD (Mamba2): x [b,s,h,p], dt [b,s,h] (post-softplus),
    A [h] (negative), B/C [b,s,g,n] broadcast over heads. Returns y
    [b,s,h,p] (D-residual applied by the caller)."""
    b, s, h, p = x.shape
    g, n = B.shape[2], B.shape[3]
    rep = h // g
    pad = (-s) % chunk
    if pad:
        x = torch.nn.functional.pad(x, (0, 0, 0, 0, 0, pad))
        dt = torch.nn.functional.pad(dt, (0, 0, 0, pad))
        B = torch.nn.functional.pad(B, (0, 0, 0, 0, 0, pad))
        C = torch.nn.functional.pad(C, (0, 0, 0, 0, 0, pad))
    S = x.shape[1]
    nc = S // chunk
    xc = x.view(b, nc, chunk, h, p)
    dtc = dt.view(b, nc, chunk, h)
    Bc = B.view(b, nc, chunk, g, n).repeat_interleave(rep, dim=3)   # [b,nc,L,h,n]
    Cc = C.view(b, nc, chunk, g, n).repeat_interleave(rep, dim=3)
    a = (dtc.float() * A.float().view(1, 1, 1, h)).permute(0, 3, 1, 2)  # [b,h,nc,L]

    decay = _segsum_decay(a)                                        # [b,h,nc,L,L]
    # within-chunk: y_i += sum_j C_i.B_j decay_ij dt_j x_j
    cb = torch.einsum("bclhn,bcjhn->bhclj", Cc.float(), Bc.float()) # [b,h,nc,L,L]
    att = cb * decay * dtc.float().permute(0, 3, 1, 2)[:, :, :, None, :]
    y = torch.einsum("bhclj,bcjhp->bclhp", att, xc.float())

    # chunk states: S_c = sum_j exp(cs_L - cs_j) dt_j B_j x_j^T  [b,h,nc,n,p]
    cs = a.cumsum(-1)
    edecay = torch.exp(cs[..., -1:] - cs)                           # [b,h,nc,L]
    w = edecay * dtc.float().permute(0, 3, 1, 2)
    states = torch.einsum("bhcl,bclhn,bclhp->bhcnp", w, Bc.float(), xc.float())

    # inter-chunk recurrence
    tot = torch.exp(cs[..., -1])                                    # [b,h,nc]
    run = torch.zeros(b, h, n, p, device=x.device, dtype=torch.float32)
    prev = []
    for c in range(nc):
        prev.append(run)
        run = run * tot[:, :, c, None, None] + states[:, :, c]
    prev = torch.stack(prev, dim=2)                                 # [b,h,nc,n,p]
    # y_i += C_i . prev * exp(cs_i)
    inner = torch.exp(cs)                                           # [b,h,nc,L]
    y = y + torch.einsum("bclhn,bhcnp,bhcl->bclhp", Cc.float(), prev, inner)
    y = y.reshape(b, S, h, p)[:, :s]
    return y


class GatedRMSNorm(nn.Module):
    """Group-wise RMSNorm(y * silu(z)) — Mamba2's gated norm (group size =
    intermediate / n_groups, matching Zamba2RMSNormGated semantics)."""

    def __init__(self, size: int, eps: float, group_size: int | None = None):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(size))
        self.eps = eps
        self.group_size = group_size or size

    def forward(self, y: torch.Tensor, z: torch.Tensor | None) -> torch.Tensor:
        dt = y.dtype
        y = y.float()
        if z is not None:
            y = y * torch.nn.functional.silu(z.float())
        *lead, last = y.shape
        yg = y.view(*lead, last // self.group_size, self.group_size)
        v = yg.pow(2).mean(-1, keepdim=True)
        yg = yg * torch.rsqrt(v + self.eps)
        return self.weight * yg.reshape(*lead, last).to(dt)


class Mamba2Mixer(nn.Module):
    """Standalone Mamba2 mixer — shared by every hybrid family (Nemotron-H,
    Bamba, and future Zamba2/FalconH1-style stacks)."""

    def __init__(self, hidden_size: int, n_heads: int, head_dim: int,
                 state_size: int, n_groups: int, conv_kernel: int = 4,
                 chunk_size: int = 128, eps: float = 1e-5,
                 use_bias: bool = False, use_conv_bias: bool = True,
                 norm_group_size: int | None = None,
                 time_step_limit: tuple = (0.0, float("inf")),
                 gate_mode: str = "norm_gate", in_scale: float = 1.0,
                 zxbcdt_multipliers: tuple | None = None):
        """gate_mode: "norm_gate" = GroupRMSNorm(y*silu(z)) (Mamba2/NemotronH/
        Bamba); "norm_then_gate" = GroupRMSNorm(y)*silu(z) (FalconH1
        norm_before_gate); "silu_only" = y*silu(z), no norm weight (FalconH1
        mamba_rms_norm=False). in_scale / zxbcdt_multipliers are FalconH1's
        muP scalars (ssm_in_multiplier; per-section z,x,B,C,dt scales applied
        to the in_proj output before the conv)."""
        super().__init__()
        inter = n_heads * head_dim
        self.n_heads, self.head_dim = n_heads, head_dim
        self.state_size, self.n_groups = state_size, n_groups
        self.chunk_size = chunk_size
        self.time_step_limit = time_step_limit
        self.inter = inter
        self.conv_dim = inter + 2 * n_groups * state_size
        self.in_proj = nn.Linear(hidden_size, inter + self.conv_dim + n_heads,
                                 bias=use_bias)
        self.conv1d = nn.Conv1d(self.conv_dim, self.conv_dim, conv_kernel,
                                groups=self.conv_dim, padding=conv_kernel - 1,
                                bias=use_conv_bias)
        self.dt_bias = nn.Parameter(torch.ones(n_heads))
        self.A_log = nn.Parameter(torch.zeros(n_heads))
        self.D = nn.Parameter(torch.ones(n_heads))
        self.gate_mode = gate_mode
        self.in_scale = in_scale
        self.zxbcdt_multipliers = (tuple(zxbcdt_multipliers)
                                   if zxbcdt_multipliers else None)
        if gate_mode == "silu_only":
            self.norm = None
        else:
            self.norm = GatedRMSNorm(inter, eps, group_size=norm_group_size)
        self.out_proj = nn.Linear(inter, hidden_size, bias=use_bias)

    def forward(self, h: torch.Tensor) -> torch.Tensor:
        b, s, _ = h.shape
        inter = self.inter
        if self.in_scale != 1.0:
            h = h * self.in_scale
        z, xBC, dt = self.in_proj(h).split(
            [inter, self.conv_dim, self.n_heads], dim=-1)
        if self.zxbcdt_multipliers is not None:
            m = self.zxbcdt_multipliers
            gts = self.n_groups * self.state_size
            z = z * m[0]
            dt = dt * m[4]
            xBC = torch.cat([xBC[..., :inter] * m[1],
                             xBC[..., inter:inter + gts] * m[2],
                             xBC[..., inter + gts:] * m[3]], dim=-1)
        xBC = torch.nn.functional.silu(
            self.conv1d(xBC.transpose(1, 2))[..., :s].transpose(1, 2))
        x, B, C = xBC.split(
            [inter, self.n_groups * self.state_size,
             self.n_groups * self.state_size], dim=-1)
        dt = torch.nn.functional.softplus(dt.float() + self.dt_bias.float())
        lo, hi = self.time_step_limit
        if lo > 0 or hi != float("inf"):
            dt = dt.clamp(min=lo, max=hi)
        A = -torch.exp(self.A_log.float())
        y = mamba2_chunked_scan(
            x.view(b, s, self.n_heads, self.head_dim), dt, A,
            B.view(b, s, self.n_groups, self.state_size),
            C.view(b, s, self.n_groups, self.state_size), self.chunk_size)
        y = y + self.D.float().view(1, 1, -1, 1) * \
            x.view(b, s, self.n_heads, self.head_dim).float()
        y = y.to(h.dtype).reshape(b, s, inter)
        if self.gate_mode == "silu_only":
            y = y * torch.nn.functional.silu(z.float()).to(y.dtype)
        elif self.gate_mode == "norm_then_gate":
            y = (self.norm(y, None) *
                 torch.nn.functional.silu(z.float())).to(y.dtype)
        else:
            y = self.norm(y, z)
        return self.out_proj(y)


class NemotronHAttentionMixer(nn.Module):
    """GQA attention, NO rope (Nemotron-H attention layers are NoPE)."""

    def __init__(self, cfg: NemotronHConfig, backend: BackendConfig):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.head_dim = D
        b = cfg.attention_bias
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=b)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=b)
        self.backend = backend

    def forward(self, h: torch.Tensor) -> torch.Tensor:
        B, S, _ = h.shape
        q = self.q_proj(h).view(B, S, -1, self.head_dim)
        k = self.k_proj(h).view(B, S, -1, self.head_dim)
        v = self.v_proj(h).view(B, S, -1, self.head_dim)
        o = flash_attention(q, k, v, causal=True, backend=self.backend.attn)
        return self.o_proj(o.reshape(B, S, -1))


class NemotronHMLPMixer(nn.Module):
    def __init__(self, cfg: NemotronHConfig):
        super().__init__()
        self.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=cfg.mlp_bias)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=cfg.mlp_bias)

    def forward(self, x):
        return self.down_proj(torch.nn.functional.relu(self.up_proj(x)).square())


class NemotronHBlock(nn.Module):
    def __init__(self, cfg: NemotronHConfig, backend: BackendConfig, kind: str):
        super().__init__()
        self.norm = RMSNorm(cfg.hidden_size, cfg.layer_norm_epsilon, backend.rms_norm)
        if kind == "M":
            self.mixer = Mamba2Mixer(
                cfg.hidden_size, cfg.mamba_num_heads, cfg.mamba_head_dim,
                cfg.ssm_state_size, cfg.n_groups, cfg.conv_kernel,
                cfg.chunk_size, cfg.layer_norm_epsilon, cfg.use_bias,
                cfg.use_conv_bias,
                norm_group_size=cfg.mamba_intermediate // cfg.n_groups,
                time_step_limit=cfg.time_step_limit)
        elif kind == "*":
            self.mixer = NemotronHAttentionMixer(cfg, backend)
        elif kind == "-":
            self.mixer = NemotronHMLPMixer(cfg)
        else:
            raise ValueError(f"unsupported hybrid layer kind '{kind}' (M/*/-)")

    def forward(self, x):
        return x + self.mixer(self.norm(x))


class NemotronHModel(nn.Module):
    def __init__(self, cfg: NemotronHConfig, backend: BackendConfig):
        super().__init__()
        pat = cfg.hybrid_override_pattern
        assert len(pat) == cfg.num_hidden_layers, \
            f"pattern length {len(pat)} != num_hidden_layers {cfg.num_hidden_layers}"
        self.embeddings = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            NemotronHBlock(cfg, backend, kind) for kind in pat)
        self.norm_f = RMSNorm(cfg.hidden_size, cfg.layer_norm_epsilon, backend.rms_norm)

    def forward(self, ids):
        x = self.embeddings(ids)
        for layer in self.layers:
            x = layer(x)
        return self.norm_f(x)


class NemotronHForCausalLM(nn.Module):
    hf_architectures = ("NemotronHForCausalLM",)
    config_class = NemotronHConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> NemotronHConfig:
        return NemotronHConfig.from_hf_config(hf_cfg)

    def __init__(self, config: NemotronHConfig | dict, backend=None):
        super().__init__()
        cfg = config if isinstance(config, NemotronHConfig) else NemotronHConfig(**dict(config))
        self.config = cfg
        bk = BackendConfig.resolve(backend, "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.head_dim)
        self.model = NemotronHModel(cfg, bk)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embeddings.weight
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
        std = cfg.initializer_range
        for m in self.modules():
            if isinstance(m, (nn.Linear, nn.Conv1d)):
                nn.init.normal_(m.weight, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, (RMSNorm, GatedRMSNorm)):
                nn.init.ones_(m.weight)
            elif isinstance(m, Mamba2Mixer):
                nn.init.ones_(m.dt_bias)
                nn.init.zeros_(m.A_log)
                nn.init.ones_(m.D)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embeddings.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
