"""OPT causal LM, MI355X-native.

Reference behavior: covered by the reference's HF model path; here native.
LEARNED positional embeddings (offset +2 in the table), biased MHA +
relu MLP (fc1/fc2), biased LayerNorms, pre-LN (``do_layer_norm_before``)
or post-LN blocks, optional ``word_embed_proj_dim`` in/out projections,
tied head. HF keys match OPTForCausalLM (parity-tested). sdpa attention.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class OPTConfig:
    vocab_size: int = 50272
    hidden_size: int = 768
    ffn_dim: int = 3072
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    max_position_embeddings: int = 2048
    word_embed_proj_dim: int | None = None
    do_layer_norm_before: bool = True
    enable_bias: bool = True
    layer_norm_elementwise_affine: bool = True
    tie_word_embeddings: bool = True
    initializer_range: float = 0.02

    def __post_init__(self):
        if self.word_embed_proj_dim is None:
            self.word_embed_proj_dim = self.hidden_size

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def from_hf_config(cls, hf: Any) -> "OPTConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        g = hf.get
        return cls(
            vocab_size=g("vocab_size", 50272),
            hidden_size=g("hidden_size", 768),
            ffn_dim=g("ffn_dim", 3072),
            num_hidden_layers=g("num_hidden_layers", 12),
            num_attention_heads=g("num_attention_heads", 12),
            max_position_embeddings=g("max_position_embeddings", 2048),
            word_embed_proj_dim=g("word_embed_proj_dim"),
            do_layer_norm_before=g("do_layer_norm_before", True),
            enable_bias=g("enable_bias", True),
            tie_word_embeddings=g("tie_word_embeddings", True),
        )


class OPTLayer(nn.Module):
    def __init__(self, cfg: OPTConfig):
        super().__init__()
        H, D, E = cfg.num_attention_heads, cfg.head_dim, cfg.hidden_size
        self.H, self.D = H, D
        self.pre_ln = cfg.do_layer_norm_before
        b = cfg.enable_bias
        attn = nn.Module()
        attn.q_proj = nn.Linear(E, E, bias=b)
        attn.k_proj = nn.Linear(E, E, bias=b)
        attn.v_proj = nn.Linear(E, E, bias=b)
        attn.out_proj = nn.Linear(E, E, bias=b)
        self.self_attn = attn
        self.self_attn_layer_norm = nn.LayerNorm(E)
        self.fc1 = nn.Linear(E, cfg.ffn_dim, bias=b)
        self.fc2 = nn.Linear(cfg.ffn_dim, E, bias=b)
        self.final_layer_norm = nn.LayerNorm(E)

    def forward(self, x):
        B, S, E = x.shape
        h = self.self_attn_layer_norm(x) if self.pre_ln else x
        a = self.self_attn
        q = a.q_proj(h).view(B, S, self.H, self.D).transpose(1, 2)
        k = a.k_proj(h).view(B, S, self.H, self.D).transpose(1, 2)
        v = a.v_proj(h).view(B, S, self.H, self.D).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v, is_causal=True)
        x = x + a.out_proj(o.transpose(1, 2).reshape(B, S, E))
        if not self.pre_ln:
            x = self.self_attn_layer_norm(x)
        h = self.final_layer_norm(x) if self.pre_ln else x
        x = x + self.fc2(F.relu(self.fc1(h)))
        if not self.pre_ln:
            x = self.final_layer_norm(x)
        return x


class OPTForCausalLM(nn.Module):
    hf_architectures = ("OPTForCausalLM",)
    config_class = OPTConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> OPTConfig:
        return OPTConfig.from_hf_config(hf_cfg)

    def __init__(self, config: OPTConfig | dict, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = OPTConfig(**config)
        self.config = config
        E, P = config.hidden_size, config.word_embed_proj_dim
        dec = nn.Module()
        dec.embed_tokens = nn.Embedding(config.vocab_size, P)
        # HF reserves 2 extra rows; lookups are offset by +2
        dec.embed_positions = nn.Embedding(config.max_position_embeddings + 2, E)
        if P != E:
            dec.project_in = nn.Linear(P, E, bias=False)
            dec.project_out = nn.Linear(E, P, bias=False)
        dec.layers = nn.ModuleList(OPTLayer(config)
                                   for _ in range(config.num_hidden_layers))
        if config.do_layer_norm_before:
            dec.final_layer_norm = nn.LayerNorm(E)
        outer = nn.Module()
        outer.decoder = dec
        self.model = outer
        self.lm_head = nn.Linear(P, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = dec.embed_tokens.weight
        self.loss_fn = None

    def forward(self, input_ids, labels=None, position_ids=None, **_: Any):
        d = self.model.decoder
        B, S = input_ids.shape
        if position_ids is None:
            position_ids = torch.arange(S, device=input_ids.device)
        else:
            position_ids = position_ids[0]
        x = d.embed_tokens(input_ids)
        if hasattr(d, "project_in"):
            x = d.project_in(x)
        x = x + d.embed_positions(position_ids + 2)[None]
        for layer in d.layers:
            x = layer(x)
        if hasattr(d, "final_layer_norm"):
            x = d.final_layer_norm(x)
        if hasattr(d, "project_out"):
            x = d.project_out(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(x, self.lm_head.weight, labels)
        return self.lm_head(x)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
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
            self.lm_head.weight = self.model.decoder.embed_tokens.weight
