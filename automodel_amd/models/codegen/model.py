"""CodeGen causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
GPT-J-style parallel-residual blocks (shared ln_1) but with a fused
``qkv_proj`` laid out in mp_num=4 groups of [q, v, k] slices per group,
partial pair-INTERLEAVED rotary (rotate_every_two, de-interleave trick),
tanh-GELU MLP, biased lm_head. HF keys match CodeGenForCausalLM
(parity-tested). Attention rides sdpa.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.ops.rope import build_rope_cache

_MP_NUM = 4   # HF CodeGen hard-codes the TPU-v4 sharding factor


@dataclass
class CodeGenConfig:
    vocab_size: int = 50400
    n_embd: int = 4096
    n_inner: int | None = None
    n_layer: int = 28
    n_head: int = 16
    rotary_dim: int = 64
    max_position_embeddings: int = 2048
    layer_norm_epsilon: float = 1e-5
    tie_word_embeddings: bool = False
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.n_embd // self.n_head

    @property
    def hidden_size(self):
        return self.n_embd

    @property
    def num_hidden_layers(self):
        return self.n_layer

    @property
    def inner_dim(self):
        return self.n_inner if self.n_inner is not None else 4 * self.n_embd

    @classmethod
    def from_hf_config(cls, hf: Any) -> "CodeGenConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        return cls(
            vocab_size=g("vocab_size", 50400),
            n_embd=g("n_embd", 4096),
            n_inner=g("n_inner"),
            n_layer=g("n_layer", 28),
            n_head=g("n_head", 16),
            rotary_dim=g("rotary_dim", 64),
            max_position_embeddings=g("max_position_embeddings",
                                      g("n_positions", 2048)),
            layer_norm_epsilon=g("layer_norm_epsilon", 1e-5),
            tie_word_embeddings=g("tie_word_embeddings", False),
        )


class CodeGenBlock(nn.Module):
    def __init__(self, cfg: CodeGenConfig):
        super().__init__()
        E, H, D = cfg.n_embd, cfg.n_head, cfg.head_dim
        self.H, self.D = H, D
        self.rot = cfg.rotary_dim
        self.ln_1 = nn.LayerNorm(E, eps=cfg.layer_norm_epsilon)
        attn = nn.Module()
        attn.qkv_proj = nn.Linear(E, 3 * E, bias=False)
        attn.out_proj = nn.Linear(E, E, bias=False)
        self.attn = attn
        mlp = nn.Module()
        mlp.fc_in = nn.Linear(E, cfg.inner_dim, bias=True)
        mlp.fc_out = nn.Linear(cfg.inner_dim, E, bias=True)
        self.mlp = mlp

    @staticmethod
    def _rot_half(t, cos, sin):
        t1, t2 = t.chunk(2, dim=-1)
        rh = torch.cat([-t2, t1], dim=-1)
        return t * cos + rh * sin

    def _rope(self, t, cos, sin):
        B, S, Hn, _ = t.shape
        r = self.rot
        tr = t[..., :r].reshape(B, S, Hn, r // 2, 2).transpose(-1, -2) \
            .reshape(B, S, Hn, r)
        return torch.cat([self._rot_half(tr, cos, sin), t[..., r:]], dim=-1)

    def forward(self, x, cos, sin):
        B, S, E = x.shape
        h = self.ln_1(x)
        # mp_num groups, each holding [q, v, k] slices of E/mp_num
        local = E // _MP_NUM
        qkv = self.attn.qkv_proj(h).view(B, S, _MP_NUM, 3 * local)
        q, v, k = torch.split(qkv, local, dim=-1)      # NOTE: q, v, k order
        hp = self.H // _MP_NUM
        q = q.reshape(B, S, self.H, self.D)
        v = v.reshape(B, S, self.H, self.D)
        k = k.reshape(B, S, self.H, self.D)
        q = self._rope(q, cos, sin)
        k = self._rope(k, cos, sin)
        o = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True)
        attn_out = self.attn.out_proj(o.transpose(1, 2).reshape(B, S, E))
        mlp_out = self.mlp.fc_out(F.gelu(self.mlp.fc_in(h), approximate="tanh"))
        return x + attn_out + mlp_out


class CodeGenForCausalLM(nn.Module):
    hf_architectures = ("CodeGenForCausalLM",)
    config_class = CodeGenConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> CodeGenConfig:
        return CodeGenConfig.from_hf_config(hf_cfg)

    def __init__(self, config: CodeGenConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = CodeGenConfig(**config)
        self.config = config
        t = nn.Module()
        t.wte = nn.Embedding(config.vocab_size, config.n_embd)
        t.h = nn.ModuleList(CodeGenBlock(config) for _ in range(config.n_layer))
        t.ln_f = nn.LayerNorm(config.n_embd, eps=config.layer_norm_epsilon)
        cos, sin = build_rope_cache(config.rotary_dim,
                                    config.max_position_embeddings, 10000.0)
        t.register_buffer("rope_cos", cos, persistent=False)
        t.register_buffer("rope_sin", sin, persistent=False)
        self.transformer = t
        self.lm_head = nn.Linear(config.n_embd, config.vocab_size, bias=True)
        if config.tie_word_embeddings:
            self.lm_head.weight = t.wte.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None, **_: Any):
        t = self.transformer
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = t.rope_cos[:S], t.rope_sin[:S]
        else:
            cos, sin = t.rope_cos[position_ids[0]], t.rope_sin[position_ids[0]]
        cos = cos.float()[None, :, None, :]
        sin = sin.float()[None, :, None, :]
        x = t.wte(input_ids)
        for block in t.h:
            x = block(x, cos, sin)
        hidden = t.ln_f(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            cos, sin = build_rope_cache(self.config.rotary_dim,
                                        self.config.max_position_embeddings,
                                        10000.0)
            self.transformer.rope_cos.copy_(cos.to(self.transformer.rope_cos.device))
            self.transformer.rope_sin.copy_(sin.to(self.transformer.rope_sin.device))
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, nn.LayerNorm):
                nn.init.ones_(mod.weight)
                nn.init.zeros_(mod.bias)
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.transformer.wte.weight
