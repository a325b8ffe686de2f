"""Mistral-3 VLM: Pixtral vision tower + patch-merger projector + mistral text.

Reference behavior: nemo_automodel/components/models/mistral3_vlm (and
mistral4). Implemented directly against the public architecture:

  * Pixtral ViT: bias-free conv patch embed, RMSNorm pre/attention/ffn
    norms, bias-free q/k/v/o and SwiGLU FFN, 2-D rotary whose angle table
    interleaves h-frequencies (even slots) and w-frequencies (odd slots)
    indexed by position h*max_w + w, block-diagonal attention per image;
  * projector: RMSNorm -> 2x2 unfold patch merger (bias-free linear) ->
    linear/GELU/linear (bias per config.multimodal_projector_bias);
  * text: the llama/mistral stack, image tokens spliced at
    image_token_index.

HF keys match Mistral3ForConditionalGeneration (parity-tested).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.llama.model import LlamaConfig, LlamaDecoderLayer
from automodel_amd.ops.rms_norm import RMSNorm, rms_norm_ref
from automodel_amd.ops.rope import build_rope_cache


@dataclass
class PixtralVisionConfig:
    hidden_size: int = 1024
    intermediate_size: int = 4096
    num_hidden_layers: int = 24
    num_attention_heads: int = 16
    image_size: int = 1540
    patch_size: int = 14
    num_channels: int = 3
    rope_theta: float = 10000.0
    head_dim: int | None = None

    def __post_init__(self):
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_attention_heads


@dataclass
class Mistral3Config:
    text: LlamaConfig = field(default_factory=LlamaConfig)
    vision: PixtralVisionConfig = field(default_factory=PixtralVisionConfig)
    image_token_id: int = 10
    spatial_merge_size: int = 2
    multimodal_projector_bias: bool = False

    def __post_init__(self):
        if isinstance(self.text, dict):
            self.text = LlamaConfig(**self.text)
        if isinstance(self.vision, dict):
            self.vision = PixtralVisionConfig(**self.vision)

    @property
    def vocab_size(self):
        return self.text.vocab_size

    @property
    def num_hidden_layers(self):
        return self.text.num_hidden_layers

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Mistral3Config":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        if "text" in hf and "vision" in hf:   # our own exported config.json
            import dataclasses as _dc

            keys = {f.name for f in _dc.fields(cls)}
            return cls(**{k: v for k, v in hf.items() if k in keys})
        tc = hf.get("text_config", hf.get("text", {}))
        vc = hf.get("vision_config", hf.get("vision", {}))
        vrp = vc.get("rope_parameters") or {}
        return cls(
            text=LlamaConfig.from_hf_config(
                dict(tc, architectures=["MistralForCausalLM"])),
            vision=PixtralVisionConfig(
                hidden_size=vc.get("hidden_size", 1024),
                intermediate_size=vc.get("intermediate_size", 4096),
                num_hidden_layers=vc.get("num_hidden_layers", 24),
                num_attention_heads=vc.get("num_attention_heads", 16),
                image_size=vc.get("image_size", 1540),
                patch_size=vc.get("patch_size", 14),
                num_channels=vc.get("num_channels", 3),
                rope_theta=vrp.get("rope_theta", vc.get("rope_theta", 10000.0)),
                head_dim=vc.get("head_dim"),
            ),
            image_token_id=hf.get("image_token_id",
                                  hf.get("image_token_index", 10)),
            spatial_merge_size=hf.get("spatial_merge_size", 2),
            multimodal_projector_bias=hf.get("multimodal_projector_bias", False),
        )


class _RMS(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        return rms_norm_ref(x, self.weight, self.eps)


class PixtralLayer(nn.Module):
    def __init__(self, cfg: PixtralVisionConfig):
        super().__init__()
        H = cfg.num_attention_heads
        self.num_heads, self.head_dim = H, cfg.head_dim
        self.attention_norm = _RMS(cfg.hidden_size)
        self.ffn_norm = _RMS(cfg.hidden_size)
        attn = nn.Module()
        attn.q_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=False)
        attn.k_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=False)
        attn.v_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=False)
        attn.o_proj = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=False)
        self.attention = attn
        ff = nn.Module()
        ff.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        ff.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        ff.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)
        self.feed_forward = ff

    def forward(self, x, cos, sin):
        h = self.attention_norm(x)
        N = h.shape[0]
        a = self.attention
        q = a.q_proj(h).view(N, self.num_heads, -1)
        k = a.k_proj(h).view(N, self.num_heads, -1)
        v = a.v_proj(h).view(N, self.num_heads, -1)

        def rot(t):
            tf = t.float()
            t1, t2 = tf.chunk(2, dim=-1)
            rh = torch.cat([-t2, t1], dim=-1)
            return (tf * cos[:, None, :] + rh * sin[:, None, :]).to(t.dtype)

        q, k = rot(q), rot(k)
        o = F.scaled_dot_product_attention(
            q.transpose(0, 1)[None], k.transpose(0, 1)[None],
            v.transpose(0, 1)[None])
        x = x + a.o_proj(o[0].transpose(0, 1).reshape(N, -1))
        h = self.ffn_norm(x)
        g = self.feed_forward.gate_proj(h)
        return x + self.feed_forward.down_proj(
            F.silu(g) * self.feed_forward.up_proj(h))


class PixtralVisionTower(nn.Module):
    def __init__(self, cfg: PixtralVisionConfig):
        super().__init__()
        self.cfg = cfg
        self.patch_conv = nn.Conv2d(cfg.num_channels, cfg.hidden_size,
                                    kernel_size=cfg.patch_size,
                                    stride=cfg.patch_size, bias=False)
        self.ln_pre = _RMS(cfg.hidden_size)
        t = nn.Module()
        t.layers = nn.ModuleList(PixtralLayer(cfg)
                                 for _ in range(cfg.num_hidden_layers))
        self.transformer = t
        # pixtral 2-D angle table: h rotates the even freq slots, w the odd
        max_side = cfg.image_size // cfg.patch_size
        d = cfg.head_dim
        freqs = 1.0 / (cfg.rope_theta ** (torch.arange(0, d, 2).float() / d))
        fh = torch.outer(torch.arange(max_side).float(), freqs[::2])
        fw = torch.outer(torch.arange(max_side).float(), freqs[1::2])
        angles = torch.cat([
            fh[:, None, :].repeat(1, max_side, 1),
            fw[None, :, :].repeat(max_side, 1, 1),
        ], dim=-1).reshape(-1, d // 2)
        angles = torch.cat([angles, angles], dim=-1)
        self.register_buffer("rope_cos", angles.cos(), persistent=False)
        self.register_buffer("rope_sin", angles.sin(), persistent=False)
        self.max_side = max_side

    def forward(self, pixel_values: torch.Tensor) -> torch.Tensor:
        """[B, C, H, W] (full square images) -> [total_patches, hidden]."""
        emb = self.patch_conv(pixel_values.to(self.patch_conv.weight.dtype))
        B, Hd, gh, gw = emb.shape
        x = emb.flatten(2).transpose(1, 2).reshape(-1, Hd)
        x = self.ln_pre(x)
        hh, ww = torch.meshgrid(torch.arange(gh), torch.arange(gw), indexing="ij")
        pos = (hh * self.max_side + ww).reshape(-1).repeat(B).to(x.device)
        cos, sin = self.rope_cos[pos], self.rope_sin[pos]
        # block-diagonal per image: with equal-size images, run per image
        outs = []
        n = gh * gw
        for b in range(B):
            xb = x[b * n:(b + 1) * n]
            cb, sb = cos[b * n:(b + 1) * n], sin[b * n:(b + 1) * n]
            for layer in self.transformer.layers:
                xb = layer(xb, cb, sb)
            outs.append(xb)
        return torch.cat(outs, dim=0)


class Mistral3ForConditionalGeneration(nn.Module):
    hf_architectures = ("Mistral3ForConditionalGeneration",)
    config_class = Mistral3Config

    @staticmethod
    def config_from_hf(hf_cfg) -> Mistral3Config:
        return Mistral3Config.from_hf_config(hf_cfg)

    def __init__(self, config: Mistral3Config | dict,
                 backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = Mistral3Config(**config)
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend),
            device_type, head_dim=config.text.head_dim)
        self.config = config
        self.backend = backend
        tc = config.text
        inner = nn.Module()
        inner.vision_tower = PixtralVisionTower(config.vision)
        proj = nn.Module()
        proj.norm = _RMS(config.vision.hidden_size, eps=tc.rms_norm_eps)
        pm = nn.Module()
        pm.merging_layer = nn.Linear(
            config.vision.hidden_size * config.spatial_merge_size ** 2,
            config.vision.hidden_size, bias=False)
        proj.patch_merger = pm
        b = config.multimodal_projector_bias
        proj.linear_1 = nn.Linear(config.vision.hidden_size, tc.hidden_size, bias=b)
        proj.linear_2 = nn.Linear(tc.hidden_size, tc.hidden_size, bias=b)
        inner.multi_modal_projector = proj
        lm = nn.Module()
        lm.embed_tokens = nn.Embedding(tc.vocab_size, tc.hidden_size)
        lm.layers = nn.ModuleList(LlamaDecoderLayer(tc, backend)
                                  for _ in range(tc.num_hidden_layers))
        lm.norm = RMSNorm(tc.hidden_size, tc.rms_norm_eps, backend.rms_norm)
        cos, sin = build_rope_cache(tc.head_dim, tc.max_position_embeddings,
                                    tc.rope_theta, tc.rope_scaling)
        lm.register_buffer("rope_cos", cos, persistent=False)
        lm.register_buffer("rope_sin", sin, persistent=False)
        inner.language_model = lm
        self.model = inner
        self.lm_head = nn.Linear(tc.hidden_size, tc.vocab_size, bias=False)
        if tc.tie_word_embeddings:
            self.lm_head.weight = lm.embed_tokens.weight
        self.loss_fn = None

    def image_features(self, pixel_values: torch.Tensor) -> torch.Tensor:
        feats = self.model.vision_tower(pixel_values)     # [T, Hv]
        proj = self.model.multi_modal_projector
        feats = proj.norm(feats)
        # 2x2 unfold merge per (square, equal-size) image
        m = self.config.spatial_merge_size
        B = pixel_values.shape[0]
        g = pixel_values.shape[-1] // self.config.vision.patch_size
        d = feats.shape[-1]
        per = g * g
        merged = []
        for b in range(B):
            grid = feats[b * per:(b + 1) * per].view(g, g, d) \
                .permute(2, 0, 1).unsqueeze(0)
            u = F.unfold(grid, kernel_size=m, stride=m)
            merged.append(u.view(d * m * m, -1).t())
        feats = proj.patch_merger.merging_layer(torch.cat(merged, dim=0))
        return proj.linear_2(F.gelu(proj.linear_1(feats)))

    def forward(self, input_ids: torch.Tensor,
                pixel_values: torch.Tensor | None = None,
                labels: torch.Tensor | None = None,
                position_ids: torch.Tensor | None = None, **_: Any) -> torch.Tensor:
        lm = self.model.language_model
        x = lm.embed_tokens(input_ids)
        if pixel_values is not None:
            img = self.image_features(pixel_values)
            mask = input_ids == self.config.image_token_id
            x = x.clone()
            x[mask] = img.to(x.dtype)
        S = input_ids.shape[1]
        if position_ids is None:
            cos, sin = lm.rope_cos[:S], lm.rope_sin[:S]
        else:
            cos, sin = lm.rope_cos[position_ids[0]], lm.rope_sin[position_ids[0]]
        cos, sin = cos.float(), sin.float()
        for layer in lm.layers:
            x = layer(x, cos, sin)
        hidden = lm.norm(x)
        if labels is not None:
            assert self.loss_fn is not None, "set model.loss_fn before labels"
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        return self.lm_head(hidden)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            tc = self.config.text
            cos, sin = build_rope_cache(tc.head_dim, tc.max_position_embeddings,
                                        tc.rope_theta, tc.rope_scaling)
            lm = self.model.language_model
            lm.rope_cos.copy_(cos.to(lm.rope_cos.device))
            lm.rope_sin.copy_(sin.to(lm.rope_sin.device))
            vt = self.model.vision_tower
            # rebuild the 2-D angle tables
            ref = PixtralVisionTower(self.config.vision)
            vt.rope_cos.copy_(ref.rope_cos.to(vt.rope_cos.device))
            vt.rope_sin.copy_(ref.rope_sin.to(vt.rope_sin.device))
        std = 0.02
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding, nn.Conv2d)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, (RMSNorm, _RMS)):
                nn.init.ones_(mod.weight)
        if self.config.text.tie_word_embeddings:
            self.lm_head.weight = self.model.language_model.embed_tokens.weight
