"""GLM-4V (GLM-4 text + EVA-style ViT with grid-sample pos-embed), native.

Reference behavior: the public GLM-4V architecture (HF
transformers.models.glm4v) — ViT with Conv3d patch embed, RMS pre/post
norms, bicubic grid_sample position resampling (align_corners=False,
border padding, per-patch (h, w) coords), SwiGLU vision MLPs, per-image
full attention, a stride-merge Conv2d downsample into an out_hidden-wide
GELU/SwiGLU merger; text is GLM-4 (sandwich post_self_attn/post_mlp
norms, fused gate_up, qkv bias, pair-INTERLEAVED partial rotary) driven
by CHUNKED-section 3D MRoPE (apply_mrope picks section chunks mod 3).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.ops.attention import flash_attention
from automodel_amd.ops.rms_norm import RMSNorm
from automodel_amd.ops.swiglu import swiglu


@dataclass
class Glm4vVisionConfig:
    depth: int = 24
    hidden_size: int = 1536
    intermediate_size: int = 13696
    num_heads: int = 12
    in_channels: int = 3
    image_size: int = 336
    patch_size: int = 14
    temporal_patch_size: int = 2
    spatial_merge_size: int = 2
    out_hidden_size: int = 4096
    rms_norm_eps: float = 1e-5
    hidden_act: str = "silu"


@dataclass
class Glm4vTextConfig:
    vocab_size: int = 151552
    hidden_size: int = 4096
    intermediate_size: int = 13696
    num_hidden_layers: int = 40
    num_attention_heads: int = 32
    num_key_value_heads: int = 2
    head_dim: int | None = None
    rms_norm_eps: float = 1e-5
    rope_theta: float = 10000.0
    partial_rotary_factor: float = 0.5
    mrope_section: tuple = (8, 12, 12)
    attention_bias: bool = True
    max_position_embeddings: int = 32768
    tie_word_embeddings: bool = False


@dataclass
class Glm4vConfig:
    text: Glm4vTextConfig = field(default_factory=Glm4vTextConfig)
    vision: Glm4vVisionConfig = field(default_factory=Glm4vVisionConfig)
    image_token_id: int = 151343
    initializer_range: float = 0.02

    def __post_init__(self):
        if isinstance(self.text, dict):
            self.text = Glm4vTextConfig(**self.text)
        if isinstance(self.vision, dict):
            self.vision = Glm4vVisionConfig(**self.vision)

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Glm4vConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        t, v = hf.get("text_config", {}), hf.get("vision_config", {})
        rp = t.get("rope_parameters") or t.get("rope_scaling") or {}
        text = Glm4vTextConfig(
            vocab_size=t.get("vocab_size", 151552),
            hidden_size=t.get("hidden_size", 4096),
            intermediate_size=t.get("intermediate_size", 13696),
            num_hidden_layers=t.get("num_hidden_layers", 40),
            num_attention_heads=t.get("num_attention_heads", 32),
            num_key_value_heads=t.get("num_key_value_heads", 2),
            head_dim=t.get("head_dim"),
            rms_norm_eps=t.get("rms_norm_eps", 1e-5),
            rope_theta=rp.get("rope_theta", t.get("rope_theta", 10000.0)),
            partial_rotary_factor=rp.get("partial_rotary_factor", 1.0),
            mrope_section=tuple(rp.get("mrope_section", (8, 12, 12))),
            attention_bias=t.get("attention_bias", True),
            max_position_embeddings=t.get("max_position_embeddings", 32768),
            tie_word_embeddings=hf.get("tie_word_embeddings",
                                       t.get("tie_word_embeddings", False)),
        )
        vision = Glm4vVisionConfig(
            depth=v.get("depth", 24),
            hidden_size=v.get("hidden_size", 1536),
            intermediate_size=v.get("intermediate_size", 13696),
            num_heads=v.get("num_heads", 12),
            in_channels=v.get("in_channels", 3),
            image_size=v.get("image_size", 336),
            patch_size=v.get("patch_size", 14),
            temporal_patch_size=v.get("temporal_patch_size", 2),
            spatial_merge_size=v.get("spatial_merge_size", 2),
            out_hidden_size=v.get("out_hidden_size", 4096),
            rms_norm_eps=v.get("rms_norm_eps", 1e-5),
            hidden_act=v.get("hidden_act", "silu"),
        )
        return cls(text=text, vision=vision,
                   image_token_id=hf.get("image_token_id", 151343))


# ---------------------------------------------------------------- vision

class Glm4vVisionBlock(nn.Module):
    def __init__(self, cfg: Glm4vVisionConfig):
        super().__init__()
        D = cfg.hidden_size
        self.n_heads = cfg.num_heads
        self.norm1 = RMSNorm(D, cfg.rms_norm_eps, "torch")
        self.norm2 = RMSNorm(D, cfg.rms_norm_eps, "torch")
        attn = nn.Module()
        attn.qkv = nn.Linear(D, 3 * D, bias=False)
        attn.proj = nn.Linear(D, D, bias=False)
        self.attn = attn
        mlp = nn.Module()
        mlp.gate_proj = nn.Linear(D, cfg.intermediate_size, bias=False)
        mlp.up_proj = nn.Linear(D, cfg.intermediate_size, bias=False)
        mlp.down_proj = nn.Linear(cfg.intermediate_size, D, bias=False)
        self.mlp = mlp

    def forward(self, x, cos, sin, seg_lens):
        L, D = x.shape
        h = self.norm1(x)
        q, k, v = (self.attn.qkv(h).reshape(L, 3, self.n_heads, -1)
                   .permute(1, 0, 2, 3).unbind(0))
        qf, kf = q.float(), k.float()
        c, s = cos.unsqueeze(-2).float(), sin.unsqueeze(-2).float()

        def rot(t):
            half = t.shape[-1] // 2
            return torch.cat([-t[..., half:], t[..., :half]], dim=-1)

        q = ((qf * c) + (rot(qf) * s)).to(x.dtype)
        k = ((kf * c) + (rot(kf) * s)).to(x.dtype)
        outs, start = [], 0
        for ln in seg_lens:
            sl = slice(start, start + ln)
            o = F.scaled_dot_product_attention(
                q[sl].transpose(0, 1)[None], k[sl].transpose(0, 1)[None],
                v[sl].transpose(0, 1)[None])
            outs.append(o[0].transpose(0, 1).reshape(ln, D))
            start += ln
        x = x + self.attn.proj(torch.cat(outs, dim=0))
        h = self.norm2(x)
        return x + self.mlp.down_proj(swiglu(self.mlp.gate_proj(h), self.mlp.up_proj(h)))


class Glm4vVisionModel(nn.Module):
    def __init__(self, cfg: Glm4vVisionConfig):
        super().__init__()
        self.cfg = cfg
        ks = [cfg.temporal_patch_size, cfg.patch_size, cfg.patch_size]
        pe = nn.Module()
        pe.proj = nn.Conv3d(cfg.in_channels, cfg.hidden_size, kernel_size=ks,
                            stride=ks, bias=True)
        self.patch_embed = pe
        emb = nn.Module()
        n_pos = (cfg.image_size // cfg.patch_size) ** 2
        emb.position_embedding = nn.Embedding(n_pos, cfg.hidden_size)
        self.embeddings = emb
        self.post_conv_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, "torch")
        self.blocks = nn.ModuleList(Glm4vVisionBlock(cfg) for _ in range(cfg.depth))
        self.post_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, "torch")
        self.downsample = nn.Conv2d(cfg.hidden_size, cfg.out_hidden_size,
                                    kernel_size=cfg.spatial_merge_size,
                                    stride=cfg.spatial_merge_size)
        merger = nn.Module()
        Dm = cfg.out_hidden_size
        merger.proj = nn.Linear(Dm, Dm, bias=False)
        merger.post_projection_norm = nn.LayerNorm(Dm)
        merger.gate_proj = nn.Linear(Dm, cfg.intermediate_size, bias=False)
        merger.up_proj = nn.Linear(Dm, cfg.intermediate_size, bias=False)
        merger.down_proj = nn.Linear(cfg.intermediate_size, Dm, bias=False)
        self.merger = merger
        head_dim = cfg.hidden_size // cfg.num_heads
        inv = 1.0 / (10000.0 ** (torch.arange(0, head_dim // 2, 2).float()
                                 / (head_dim // 2)))
        self.register_buffer("rot_inv_freq", inv, persistent=False)

    def _coords(self, grid_thw):
        m = self.cfg.spatial_merge_size
        out = []
        for t, h, w in grid_thw.tolist():
            hh, ww = torch.meshgrid(torch.arange(h), torch.arange(w), indexing="ij")
            shape = (h // m, m, w // m, m)
            hh = hh.reshape(shape).transpose(1, 2).flatten()
            ww = ww.reshape(shape).transpose(1, 2).flatten()
            out.append(torch.stack([hh, ww], dim=-1).repeat(t, 1))
        return torch.cat(out, dim=0)

    def forward(self, pixel_values: torch.Tensor, grid_thw: torch.Tensor):
        cfg = self.cfg
        x = pixel_values.view(-1, cfg.in_channels, cfg.temporal_patch_size,
                              cfg.patch_size, cfg.patch_size)
        x = self.patch_embed.proj(x.to(self.patch_embed.proj.weight.dtype))
        x = x.view(-1, cfg.hidden_size)
        x = self.post_conv_layernorm(x)

        pos = self._coords(grid_thw).to(x.device)                  # [L, 2] (h, w)
        freqs = (pos.unsqueeze(-1).float()
                 * self.rot_inv_freq.to(x.device)).flatten(1)
        emb = torch.cat([freqs, freqs], dim=-1)
        cos, sin = emb.cos(), emb.sin()

        # bicubic grid_sample pos-embed resample (align_corners=False, border)
        side = int(self.embeddings.position_embedding.weight.shape[0] ** 0.5)
        table = (self.embeddings.position_embedding.weight
                 .view(side, side, -1).permute(2, 0, 1)[None].float())
        segs = [t * h * w for t, h, w in grid_thw.tolist()]
        tgt = torch.repeat_interleave(
            grid_thw[:, 1:].float(), torch.tensor(segs), dim=0).to(x.device)
        norm_w = ((pos[:, 1].float() + 0.5) / tgt[:, 1]) * 2 - 1
        norm_h = ((pos[:, 0].float() + 0.5) / tgt[:, 0]) * 2 - 1
        grid = torch.stack((norm_w, norm_h), dim=-1)[None, :, None]
        pe = F.grid_sample(table, grid, mode="bicubic", align_corners=False,
                           padding_mode="border")[0, :, :, 0].permute(1, 0)
        x = x + pe.to(x.dtype)

        for blk in self.blocks:
            x = blk(x, cos, sin, segs)
        x = self.post_layernorm(x)
        m = cfg.spatial_merge_size
        x = x.view(-1, m, m, x.shape[-1]).permute(0, 3, 1, 2)
        x = self.downsample(x).view(-1, cfg.out_hidden_size)
        mg = self.merger
        h = F.gelu(mg.post_projection_norm(mg.proj(x)))
        return mg.down_proj(swiglu(mg.gate_proj(h), mg.up_proj(h)))


# ---------------------------------------------------------------- text

class Glm4vTextLayer(nn.Module):
    def __init__(self, cfg: Glm4vTextConfig, backend: BackendConfig):
        super().__init__()
        H, Hk = cfg.num_attention_heads, cfg.num_key_value_heads
        D = cfg.head_dim or cfg.hidden_size // H
        self.head_dim = D
        self.rot = int(D * cfg.partial_rotary_factor)
        attn = nn.Module()
        b = cfg.attention_bias
        attn.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=b)
        attn.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        attn.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        attn.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=False)
        self.self_attn = attn
        mlp = nn.Module()
        mlp.gate_up_proj = nn.Linear(cfg.hidden_size, 2 * cfg.intermediate_size,
                                     bias=False)
        mlp.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)
        self.mlp = mlp
        eps = cfg.rms_norm_eps
        self.input_layernorm = RMSNorm(cfg.hidden_size, eps, backend.rms_norm)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, eps, backend.rms_norm)
        self.post_self_attn_layernorm = RMSNorm(cfg.hidden_size, eps, backend.rms_norm)
        self.post_mlp_layernorm = RMSNorm(cfg.hidden_size, eps, backend.rms_norm)
        self.backend = backend

    @staticmethod
    def _rope_interleaved(t, cos, sin, rot):
        # glm pair-interleaved partial rope: cos/sin arrive [B, S, rot]
        c = cos[..., : rot // 2].repeat_interleave(2, dim=-1)[:, :, None]
        s = sin[..., : rot // 2].repeat_interleave(2, dim=-1)[:, :, None]
        tr, tp = t[..., :rot].float(), t[..., rot:]
        x1, x2 = tr[..., 0::2], tr[..., 1::2]
        rot_t = torch.stack((-x2, x1), dim=-1).flatten(-2)
        return torch.cat([(tr * c + rot_t * s).to(t.dtype), tp], dim=-1)

    def forward(self, x, cos, sin):
        a = self.self_attn
        B, S, _ = x.shape
        D = self.head_dim
        h = self.input_layernorm(x)
        q = a.q_proj(h).view(B, S, -1, D)
        k = a.k_proj(h).view(B, S, -1, D)
        v = a.v_proj(h).view(B, S, -1, D)
        q = self._rope_interleaved(q, cos.float(), sin.float(), self.rot)
        k = self._rope_interleaved(k, cos.float(), sin.float(), self.rot)
        o = flash_attention(q, k, v, causal=True, backend=self.backend.attn)
        h = self.post_self_attn_layernorm(a.o_proj(o.reshape(B, S, -1)))
        x = x + h
        h = self.post_attention_layernorm(x)
        gate, up = self.mlp.gate_up_proj(h).chunk(2, dim=-1)
        h = self.post_mlp_layernorm(self.mlp.down_proj(swiglu(gate, up)))
        return x + h


class Glm4vTextModel(nn.Module):
    def __init__(self, cfg: Glm4vTextConfig, backend: BackendConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            Glm4vTextLayer(cfg, backend) for _ in range(cfg.num_hidden_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        D = cfg.head_dim or cfg.hidden_size // cfg.num_attention_heads
        rot = int(D * cfg.partial_rotary_factor)
        inv = 1.0 / (cfg.rope_theta ** (torch.arange(0, rot, 2).float() / rot))
        self.register_buffer("inv_freq", inv, persistent=False)

    def _mrope(self, position_ids, dtype):
        """Chunked-section mrope: freqs [3, B, S, rot/2] -> section chunks
        taken from axis i%3 (reference apply_mrope)."""
        freqs = torch.einsum("nbs,d->nbsd", position_ids.float(),
                             self.inv_freq.float())
        chunks = freqs.split(list(self.cfg.mrope_section), dim=-1)
        out = torch.cat([c[i % 3] for i, c in enumerate(chunks)], dim=-1)
        emb = torch.cat([out, out], dim=-1)
        return emb.cos().to(dtype), emb.sin().to(dtype)

    def forward(self, embeds, position_ids):
        cos, sin = self._mrope(position_ids, torch.float32)
        x = embeds
        for layer in self.layers:
            x = layer(x, cos, sin)
        return self.norm(x)


class Glm4vForConditionalGeneration(nn.Module):
    hf_architectures = ("Glm4vForConditionalGeneration",)
    config_class = Glm4vConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> Glm4vConfig:
        return Glm4vConfig.from_hf_config(hf_cfg)

    def __init__(self, config: Glm4vConfig | dict, backend=None):
        super().__init__()
        cfg = config if isinstance(config, Glm4vConfig) else Glm4vConfig(**dict(config))
        self.config = cfg
        D = cfg.text.head_dim or cfg.text.hidden_size // cfg.text.num_attention_heads
        bk = BackendConfig.resolve(backend,
                                   "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=D)
        inner = nn.Module()
        inner.visual = Glm4vVisionModel(cfg.vision)
        inner.language_model = Glm4vTextModel(cfg.text, bk)
        self.model = inner
        self.lm_head = nn.Linear(cfg.text.hidden_size, cfg.text.vocab_size, bias=False)
        if cfg.text.tie_word_embeddings:
            self.lm_head.weight = inner.language_model.embed_tokens.weight
        self.loss_fn = None

    def get_rope_index(self, input_ids, image_grid_thw):
        cfg = self.config
        m = cfg.vision.spatial_merge_size
        B, S = input_ids.shape
        pos = torch.zeros(3, B, S, dtype=torch.long, device=input_ids.device)
        grids = iter(image_grid_thw.tolist() if image_grid_thw is not None else [])
        for b in range(B):
            is_img = (input_ids[b] == cfg.image_token_id)
            cur = 0
            i = 0
            out = []
            while i < S:
                if not bool(is_img[i]):
                    j = i
                    while j < S and not bool(is_img[j]):
                        j += 1
                    out.append(torch.arange(j - i, device=input_ids.device)
                               .view(1, -1).expand(3, -1) + cur)
                    cur += j - i
                    i = j
                else:
                    t, h, w = next(grids)
                    hm, wm = h // m, w // m
                    tt = torch.arange(t).repeat_interleave(hm * wm)
                    hh = torch.arange(hm).repeat_interleave(wm).repeat(t)
                    ww = torch.arange(wm).repeat(hm).repeat(t)
                    out.append(torch.stack([tt + cur, hh + cur, ww + cur])
                               .to(input_ids.device))
                    cur += max(h, w) // m
                    i += t * hm * wm
            pos[:, b] = torch.cat(out, dim=1)
        return pos

    def forward(self, input_ids, pixel_values=None, image_grid_thw=None,
                labels=None, position_ids=None, return_hidden=False, **_):
        cfg = self.config
        lm = self.model.language_model
        embeds = lm.embed_tokens(input_ids)
        mask = None
        if pixel_values is not None and pixel_values.numel() > 0:
            img_tokens = self.model.visual(pixel_values.to(embeds.dtype),
                                           image_grid_thw)
            mask = input_ids == cfg.image_token_id
            assert int(mask.sum()) == img_tokens.shape[0], "image slots != tokens"
            embeds = embeds.clone()
            embeds[mask] = img_tokens.to(embeds.dtype)
        if position_ids is None:
            if mask is not None:
                position_ids = self.get_rope_index(input_ids, image_grid_thw)
            else:
                S = input_ids.shape[1]
                position_ids = (torch.arange(S, device=input_ids.device)
                                .view(1, 1, -1).expand(3, input_ids.shape[0], -1))
        h = lm(embeds, position_ids)
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

    def freeze_vision_tower(self) -> None:
        for p in self.model.visual.parameters():
            p.requires_grad_(False)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
        std = self.config.initializer_range
        for m in self.modules():
            if isinstance(m, (nn.Linear, nn.Conv2d, nn.Conv3d)):
                nn.init.normal_(m.weight, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, nn.LayerNorm):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
            elif type(m).__name__ == "RMSNorm":
                nn.init.ones_(m.weight)
        if self.config.text.tie_word_embeddings:
            self.lm_head.weight = self.model.language_model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
