"""Qwen3-VL (DeepStack ViT + interleaved-MRoPE Qwen3 LM), MI355X-native.

Reference behavior: the public Qwen3-VL architecture (HF
transformers.models.qwen3_vl) — ViT with Conv3d patch embed, learned
position table bilinearly resampled (align_corners) to each image grid in
spatial-merge-block order, 2-axis (h,w) rotary, full per-image attention,
a 2x2 patch merger, and DeepStack (arXiv 2406.04334): intermediate ViT
features from ``deepstack_visual_indexes`` are merged (post-shuffle norm)
and ADDED to the hidden states of the first k decoder layers at visual
positions. Text side is Qwen3 (per-head q/k RMSNorm) with INTERLEAVED
3D MRoPE ([THWTHW...TT] frequency layout, modeling_qwen3_vl.py:372) and
get_rope_index-style text/image position bookkeeping.
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


def _gelu_tanh(x):
    return F.gelu(x, approximate="tanh")


@dataclass
class Qwen3VLVisionConfig:
    depth: int = 27
    hidden_size: int = 1152
    intermediate_size: int = 4304
    num_heads: int = 16
    in_channels: int = 3
    patch_size: int = 16
    temporal_patch_size: int = 2
    spatial_merge_size: int = 2
    out_hidden_size: int = 3584
    num_position_embeddings: int = 2304
    deepstack_visual_indexes: list = field(default_factory=lambda: [8, 16, 24])
    hidden_act: str = "gelu_pytorch_tanh"


@dataclass
class Qwen3VLTextConfig:
    vocab_size: int = 151936
    hidden_size: int = 2048
    intermediate_size: int = 6144
    num_hidden_layers: int = 28
    num_attention_heads: int = 16
    num_key_value_heads: int = 8
    head_dim: int = 128
    rms_norm_eps: float = 1e-6
    rope_theta: float = 5e6
    mrope_section: tuple = (24, 20, 20)
    attention_bias: bool = False
    max_position_embeddings: int = 128000
    tie_word_embeddings: bool = False
    # MoE variant (Qwen3VLMoe): softmax-topk-renorm routing, fused experts
    num_experts: int = 0
    num_experts_per_tok: int = 8
    moe_intermediate_size: int = 768
    decoder_sparse_step: int = 1
    mlp_only_layers: tuple = ()


@dataclass
class Qwen3VLConfig:
    text: Qwen3VLTextConfig = field(default_factory=Qwen3VLTextConfig)
    vision: Qwen3VLVisionConfig = field(default_factory=Qwen3VLVisionConfig)
    image_token_id: int = 151655
    initializer_range: float = 0.02

    def __post_init__(self):
        if isinstance(self.text, dict):
            self.text = Qwen3VLTextConfig(**self.text)
        if isinstance(self.vision, dict):
            self.vision = Qwen3VLVisionConfig(**self.vision)

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Qwen3VLConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        t, v = hf.get("text_config", {}), hf.get("vision_config", {})
        rp = t.get("rope_parameters") or t.get("rope_scaling") or {}
        text = Qwen3VLTextConfig(
            vocab_size=t.get("vocab_size", 151936),
            hidden_size=t.get("hidden_size", 2048),
            intermediate_size=t.get("intermediate_size", 6144),
            num_hidden_layers=t.get("num_hidden_layers", 28),
            num_attention_heads=t.get("num_attention_heads", 16),
            num_key_value_heads=t.get("num_key_value_heads", 8),
            head_dim=t.get("head_dim", 128),
            rms_norm_eps=t.get("rms_norm_eps", 1e-6),
            rope_theta=rp.get("rope_theta", t.get("rope_theta", 5e6)),
            mrope_section=tuple(rp.get("mrope_section", (24, 20, 20))),
            attention_bias=t.get("attention_bias", False),
            max_position_embeddings=t.get("max_position_embeddings", 128000),
            tie_word_embeddings=hf.get("tie_word_embeddings",
                                       t.get("tie_word_embeddings", False)),
            num_experts=t.get("num_experts") or t.get("num_local_experts") or 0,
            num_experts_per_tok=t.get("num_experts_per_tok", 8),
            moe_intermediate_size=t.get("moe_intermediate_size", 768),
            decoder_sparse_step=t.get("decoder_sparse_step", 1),
            mlp_only_layers=tuple(t.get("mlp_only_layers") or ()),
        )
        vision = Qwen3VLVisionConfig(
            depth=v.get("depth", 27),
            hidden_size=v.get("hidden_size", 1152),
            intermediate_size=v.get("intermediate_size", 4304),
            num_heads=v.get("num_heads", 16),
            in_channels=v.get("in_channels", 3),
            patch_size=v.get("patch_size", 16),
            temporal_patch_size=v.get("temporal_patch_size", 2),
            spatial_merge_size=v.get("spatial_merge_size", 2),
            out_hidden_size=v.get("out_hidden_size", 3584),
            num_position_embeddings=v.get("num_position_embeddings", 2304),
            deepstack_visual_indexes=list(v.get("deepstack_visual_indexes", [8, 16, 24])),
            hidden_act=v.get("hidden_act", "gelu_pytorch_tanh"),
        )
        return cls(text=text, vision=vision,
                   image_token_id=hf.get("image_token_id", 151655))


# ---------------------------------------------------------------- vision

class VisionPatchMerger(nn.Module):
    def __init__(self, cfg: Qwen3VLVisionConfig, postshuffle_norm: bool):
        super().__init__()
        merged = cfg.hidden_size * cfg.spatial_merge_size ** 2
        self.merged = merged
        self.postshuffle = postshuffle_norm
        self.norm = nn.LayerNorm(merged if postshuffle_norm else cfg.hidden_size,
                                 eps=1e-6)
        self.linear_fc1 = nn.Linear(merged, merged)
        self.linear_fc2 = nn.Linear(merged, cfg.out_hidden_size)

    def forward(self, x):
        x = self.norm(x.view(-1, self.merged) if self.postshuffle else x)
        return self.linear_fc2(F.gelu(self.linear_fc1(x.view(-1, self.merged))))


class VisionBlock(nn.Module):
    def __init__(self, cfg: Qwen3VLVisionConfig):
        super().__init__()
        D = cfg.hidden_size
        self.n_heads = cfg.num_heads
        self.norm1 = nn.LayerNorm(D, eps=1e-6)
        self.norm2 = nn.LayerNorm(D, eps=1e-6)
        attn = nn.Module()
        attn.qkv = nn.Linear(D, 3 * D, bias=True)
        attn.proj = nn.Linear(D, D, bias=True)
        self.attn = attn
        mlp = nn.Module()
        mlp.linear_fc1 = nn.Linear(D, cfg.intermediate_size, bias=True)
        mlp.linear_fc2 = nn.Linear(cfg.intermediate_size, D, bias=True)
        self.mlp = mlp
        self.act = _gelu_tanh if cfg.hidden_act == "gelu_pytorch_tanh" else getattr(F, cfg.hidden_act)

    def forward(self, x, cos, sin, seg_lens):
        L, D = x.shape
        h = self.norm1(x)
        q, k, v = (self.attn.qkv(h).reshape(L, 3, self.n_heads, -1)
                   .permute(1, 0, 2, 3).unbind(0))
        # full-head rotary from the (h|w) frequency concat, fp32
        qf, kf = q.float(), k.float()
        c, s = cos.unsqueeze(-2).float(), sin.unsqueeze(-2).float()

        def rot(t):
            half = t.shape[-1] // 2
            return torch.cat([-t[..., half:], t[..., :half]], dim=-1)

        q = ((qf * c) + (rot(qf) * s)).to(x.dtype)
        k = ((kf * c) + (rot(kf) * s)).to(x.dtype)
        outs = []
        start = 0
        for ln in seg_lens:      # full attention within each image
            sl = slice(start, start + ln)
            o = F.scaled_dot_product_attention(
                q[sl].transpose(0, 1)[None], k[sl].transpose(0, 1)[None],
                v[sl].transpose(0, 1)[None])
            outs.append(o[0].transpose(0, 1).reshape(ln, D))
            start += ln
        x = x + self.attn.proj(torch.cat(outs, dim=0))
        return x + self.mlp.linear_fc2(self.act(self.mlp.linear_fc1(self.norm2(x))))


class Qwen3VLVisionModel(nn.Module):
    def __init__(self, cfg: Qwen3VLVisionConfig):
        super().__init__()
        self.cfg = cfg
        ks = [cfg.temporal_patch_size, cfg.patch_size, cfg.patch_size]
        pe = nn.Module()
        pe.proj = nn.Conv3d(cfg.in_channels, cfg.hidden_size, kernel_size=ks,
                            stride=ks, bias=True)
        self.patch_embed = pe
        self.pos_embed = nn.Embedding(cfg.num_position_embeddings, cfg.hidden_size)
        self.grid_side = int(cfg.num_position_embeddings ** 0.5)
        self.blocks = nn.ModuleList(VisionBlock(cfg) for _ in range(cfg.depth))
        self.merger = VisionPatchMerger(cfg, postshuffle_norm=False)
        self.deepstack_merger_list = nn.ModuleList(
            VisionPatchMerger(cfg, postshuffle_norm=True)
            for _ in cfg.deepstack_visual_indexes)
        head_dim = cfg.hidden_size // cfg.num_heads
        inv = 1.0 / (10000.0 ** (torch.arange(0, head_dim // 2, 2).float()
                                 / (head_dim // 2)))
        self.register_buffer("rot_inv_freq", inv, persistent=False)

    def _merge_order(self, h, w):
        m = self.cfg.spatial_merge_size
        idx = torch.arange(h * w).reshape(h // m, m, w // m, m)
        return idx.transpose(1, 2).reshape(-1)

    def _pos_embed_for(self, t, h, w, dtype):
        """Bilinear align_corners resample of the square table to (h, w),
        emitted in merge-block order, repeated t times."""
        side = self.grid_side
        table = self.pos_embed.weight.view(side, side, -1).permute(2, 0, 1)[None]
        pe = F.interpolate(table.float(), size=(h, w), mode="bilinear",
                           align_corners=True)[0].permute(1, 2, 0).reshape(h * w, -1)
        pe = pe[self._merge_order(h, w).to(pe.device)]
        return pe.repeat(t, 1).to(dtype)

    def _rot_freqs(self, t, h, w, device):
        m = self.cfg.spatial_merge_size
        hh, ww = torch.meshgrid(torch.arange(h, device=device),
                                torch.arange(w, device=device), indexing="ij")
        shape = (h // m, m, w // m, m)
        hh = hh.reshape(shape).transpose(1, 2).flatten()
        ww = ww.reshape(shape).transpose(1, 2).flatten()
        pos = torch.stack([hh, ww], dim=-1).repeat(t, 1)      # [L, 2]
        freqs = (pos.unsqueeze(-1).float() * self.rot_inv_freq.to(device)).flatten(1)
        emb = torch.cat([freqs, freqs], dim=-1)
        return emb.cos(), emb.sin()

    def forward(self, pixel_values: torch.Tensor, grid_thw: torch.Tensor):
        """pixel_values [n_patches, C*tp*ps*ps] (HF packed layout);
        grid_thw [n_images, 3]. Returns (merged tokens, deepstack list)."""
        cfg = self.cfg
        x = pixel_values.view(-1, cfg.in_channels, cfg.temporal_patch_size,
                              cfg.patch_size, cfg.patch_size)
        x = self.patch_embed.proj(x.to(self.patch_embed.proj.weight.dtype))
        x = x.view(-1, cfg.hidden_size)
        pes, coss, sins, segs = [], [], [], []
        for t, h, w in grid_thw.tolist():
            pes.append(self._pos_embed_for(t, h, w, x.dtype))
            c, s = self._rot_freqs(t, h, w, x.device)
            coss.append(c)
            sins.append(s)
            segs.append(t * h * w)
        x = x + torch.cat(pes, dim=0)
        cos, sin = torch.cat(coss, dim=0), torch.cat(sins, dim=0)
        deepstack = []
        for i, blk in enumerate(self.blocks):
            x = blk(x, cos, sin, segs)
            if i in cfg.deepstack_visual_indexes:
                j = cfg.deepstack_visual_indexes.index(i)
                deepstack.append(self.deepstack_merger_list[j](x))
        return self.merger(x), deepstack


# ---------------------------------------------------------------- text

class Qwen3VLTextAttention(nn.Module):
    def __init__(self, cfg: Qwen3VLTextConfig, backend: BackendConfig):
        super().__init__()
        H, Hk, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        self.head_dim = D
        b = cfg.attention_bias
        self.q_proj = nn.Linear(cfg.hidden_size, H * D, bias=b)
        self.k_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.v_proj = nn.Linear(cfg.hidden_size, Hk * D, bias=b)
        self.o_proj = nn.Linear(H * D, cfg.hidden_size, bias=b)
        self.q_norm = RMSNorm(D, cfg.rms_norm_eps, "torch")
        self.k_norm = RMSNorm(D, cfg.rms_norm_eps, "torch")
        self.backend = backend

    def forward(self, h, cos, sin):
        B, S, _ = h.shape
        D = self.head_dim
        q = self.q_norm(self.q_proj(h).view(B, S, -1, D))
        k = self.k_norm(self.k_proj(h).view(B, S, -1, D))
        v = self.v_proj(h).view(B, S, -1, D)

        def rot(t):
            half = t.shape[-1] // 2
            return torch.cat([-t[..., half:], t[..., :half]], dim=-1)

        c, s = cos[:, :, None].float(), sin[:, :, None].float()
        q = ((q.float() * c) + (rot(q.float()) * s)).to(h.dtype)
        k = ((k.float() * c) + (rot(k.float()) * s)).to(h.dtype)
        o = flash_attention(q, k, v, causal=True, backend=self.backend.attn)
        return self.o_proj(o.reshape(B, S, -1))


class Qwen3VLTextLayer(nn.Module):
    def __init__(self, cfg: Qwen3VLTextConfig, backend: BackendConfig,
                 layer_idx: int = 0):
        super().__init__()
        self.self_attn = Qwen3VLTextAttention(cfg, backend)
        sparse = (cfg.num_experts > 0
                  and layer_idx not in (cfg.mlp_only_layers or ())
                  and (layer_idx + 1) % cfg.decoder_sparse_step == 0)
        self.is_moe = sparse
        if sparse:
            from automodel_amd.moe.config import MoEConfig
            from automodel_amd.moe.layers import MoE

            self.mlp = MoE(cfg.hidden_size, MoEConfig(
                n_routed_experts=cfg.num_experts,
                n_activated_experts=cfg.num_experts_per_tok,
                moe_intermediate_size=cfg.moe_intermediate_size,
                norm_topk_prob=True))
        else:
            mlp = nn.Module()
            mlp.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
            mlp.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
            mlp.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)
            self.mlp = mlp
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps,
                                                backend.rms_norm)

    def forward(self, x, cos, sin):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin)
        h = self.post_attention_layernorm(x)
        if self.is_moe:
            return x + self.mlp(h)
        return x + self.mlp.down_proj(swiglu(self.mlp.gate_proj(h), self.mlp.up_proj(h)))


class Qwen3VLTextModel(nn.Module):
    def __init__(self, cfg: Qwen3VLTextConfig, backend: BackendConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            Qwen3VLTextLayer(cfg, backend, i) for i in range(cfg.num_hidden_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps, backend.rms_norm)
        D = cfg.head_dim
        inv = 1.0 / (cfg.rope_theta ** (torch.arange(0, D, 2).float() / D))
        self.register_buffer("inv_freq", inv, persistent=False)

    def _mrope(self, position_ids: torch.Tensor, dtype):
        """position_ids [3, B, S] -> interleaved cos/sin [B, S, D]."""
        inv = self.inv_freq.float()
        freqs = torch.einsum("nbs,d->nbsd", position_ids.float(), inv)  # [3,B,S,D/2]
        out = freqs[0].clone()
        ms = self.cfg.mrope_section
        for dim, offset in ((1, 1), (2, 2)):
            idx = slice(offset, ms[dim] * 3, 3)
            out[..., idx] = freqs[dim][..., idx]
        emb = torch.cat([out, out], dim=-1)
        return emb.cos().to(dtype), emb.sin().to(dtype)

    def forward(self, embeds, position_ids, visual_pos_mask=None, deepstack=None):
        cos, sin = self._mrope(position_ids, torch.float32)
        x = embeds
        for i, layer in enumerate(self.layers):
            x = layer(x, cos, sin)
            if deepstack is not None and i < len(deepstack):
                x = x.clone()
                x[visual_pos_mask] = x[visual_pos_mask] + \
                    deepstack[i].to(x.dtype)
        return self.norm(x)


class Qwen3VLForConditionalGeneration(nn.Module):
    hf_architectures = ("Qwen3VLForConditionalGeneration",)
    config_class = Qwen3VLConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> Qwen3VLConfig:
        return Qwen3VLConfig.from_hf_config(hf_cfg)

    def __init__(self, config: Qwen3VLConfig | dict, backend=None):
        super().__init__()
        cfg = config if isinstance(config, Qwen3VLConfig) else Qwen3VLConfig(**dict(config))
        self.config = cfg
        bk = BackendConfig.resolve(backend,
                                   "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.text.head_dim)
        inner = nn.Module()
        inner.visual = Qwen3VLVisionModel(cfg.vision)
        inner.language_model = Qwen3VLTextModel(cfg.text, bk)
        self.model = inner
        self.lm_head = nn.Linear(cfg.text.hidden_size, cfg.text.vocab_size, bias=False)
        if cfg.text.tie_word_embeddings:
            self.lm_head.weight = inner.language_model.embed_tokens.weight
        self.loss_fn = None

    def get_rope_index(self, input_ids, image_grid_thw):
        """Images-only 3D rope index (reference get_rope_index, text runs
        share one position; image blocks get (t+pos, h, w) grids)."""
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
                    n = j - i
                    out.append(torch.arange(n, device=input_ids.device)
                               .view(1, -1).expand(3, -1) + cur)
                    cur += n
                    i = j
                else:
                    t, h, w = next(grids)
                    hm, wm = h // m, w // m
                    tt = torch.arange(t).repeat_interleave(hm * wm)
                    hh = torch.arange(hm).repeat_interleave(wm).repeat(t)
                    ww = torch.arange(wm).repeat(hm).repeat(t)
                    blk = torch.stack([tt + cur, hh + cur, ww + cur]).to(input_ids.device)
                    out.append(blk)
                    cur += max(h, w) // m
                    i += t * hm * wm
            pos[:, b] = torch.cat(out, dim=1)
        return pos

    def forward(self, input_ids, pixel_values=None, image_grid_thw=None,
                labels=None, position_ids=None, return_hidden=False, **_):
        cfg = self.config
        lm = self.model.language_model
        embeds = lm.embed_tokens(input_ids)
        deepstack = None
        mask = None
        if pixel_values is not None and pixel_values.numel() > 0:
            img_tokens, deepstack = self.model.visual(
                pixel_values.to(embeds.dtype), image_grid_thw)
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
        h = lm(embeds, position_ids, visual_pos_mask=mask, deepstack=deepstack)
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
            if isinstance(m, (nn.Linear, nn.Conv3d)):
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
            elif type(m).__name__ == "MoE":          # Qwen3VLMoe text FFNs
                nn.init.normal_(m.gate.weight, std=std)
                nn.init.normal_(m.experts.gate_proj, std=std)
                nn.init.normal_(m.experts.up_proj, std=std)
                nn.init.normal_(m.experts.down_proj, std=std)
        if self.config.text.tie_word_embeddings:
            self.lm_head.weight = self.model.language_model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())

class Qwen3VLMoeStateDictAdapter:
    """HF fused experts.gate_up_proj [E,2I,H] <-> stacked gate/up [E,I,H]."""

    def from_hf(self, sd: dict) -> dict:
        out = {}
        for k, v in sd.items():
            if k.endswith("mlp.experts.gate_up_proj"):
                gate, up = v.chunk(2, dim=1)
                out[k.replace("gate_up_proj", "gate_proj")] = gate.contiguous()
                out[k.replace("gate_up_proj", "up_proj")] = up.contiguous()
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
            else:
                out[k] = v
        return out


class Qwen3VLMoeForConditionalGeneration(Qwen3VLForConditionalGeneration):
    """Qwen3-VL-MoE: same DeepStack vision + MRoPE text with softmax-topk-
    renorm MoE FFNs (reference transformers.models.qwen3_vl_moe)."""

    hf_architectures = ("Qwen3VLMoeForConditionalGeneration",)
    state_dict_adapter = Qwen3VLMoeStateDictAdapter
