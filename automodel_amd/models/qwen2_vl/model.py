"""Qwen2-VL: ViT vision tower + m-rope text decoder, MI355X-native.

Reference behavior: nemo_automodel's VLM families (components/models/
qwen3_vl/ etc. and recipes/vlm). Implemented directly against the public
Qwen2-VL architecture:

  * vision tower: Conv3d patch embed (temporal patch 2), 2-axis rotary over
    block-major (h, w) positions (first quarter of head_dim rotates h, the
    second w), non-causal full attention per image (fused qkv + bias),
    quick-GELU MLP, and a 2x2 spatial merger projecting into the text width;
  * text decoder: the llama/qwen2 stack (attention bias) with MULTIMODAL
    3-D rope — position_ids [3, B, S] (temporal/height/width); the three
    cos/sin variants are merged channel-wise by ``mrope_section`` into ONE
    batch-shaped table, so the decoder layers run unchanged (apply_rope
    broadcasts [B, S, D] tables);
  * image splice: <|image_pad|> token runs are replaced by merged vision
    embeddings; ``get_rope_index`` assigns text tokens sequential positions
    and image tokens their (t, h, w) grid offset by the running position.

State-dict keys match HF Qwen2VLForConditionalGeneration (parity-tested).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.llama.model import LlamaConfig, LlamaDecoderLayer
from automodel_amd.ops.rms_norm import RMSNorm


@dataclass
class VisionConfig:
    embed_dim: int = 1280
    depth: int = 32
    num_heads: int = 16
    mlp_ratio: float = 4.0
    patch_size: int = 14
    temporal_patch_size: int = 2
    spatial_merge_size: int = 2
    in_channels: int = 3
    hidden_size: int = 3584          # text width (merger output)
    # ---- qwen2.5-vl variant ----
    variant: str = "v2"              # "v2" (LN + quick-gelu) | "v2_5" (RMS + swiglu + windows)
    intermediate_size: int | None = None     # v2_5 explicit mlp width
    window_size: int = 112
    fullatt_block_indexes: tuple = (7, 15, 23, 31)
    # omni towers keep separate q/k/v linears instead of a fused qkv
    qkv_separate: bool = False


@dataclass
class Qwen2VLConfig:
    text: LlamaConfig = field(default_factory=LlamaConfig)
    vision: VisionConfig = field(default_factory=VisionConfig)
    mrope_section: tuple = (16, 24, 24)
    image_token_id: int = 151655
    vision_start_token_id: int = 151652

    def __post_init__(self):
        if isinstance(self.text, dict):
            self.text = LlamaConfig(**self.text)
        if isinstance(self.vision, dict):
            self.vision = VisionConfig(**self.vision)
        self.mrope_section = tuple(self.mrope_section)

    # recipe plumbing reads these off model.config
    @property
    def vocab_size(self):
        return self.text.vocab_size

    @property
    def num_hidden_layers(self):
        return self.text.num_hidden_layers

    @property
    def num_key_value_heads(self):
        return self.text.num_key_value_heads

    @property
    def head_dim(self):
        return self.text.head_dim

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Qwen2VLConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        if "text" in hf and "vision" in hf:   # our own exported config.json
            import dataclasses as _dc

            keys = {f.name for f in _dc.fields(cls)}
            return cls(**{k: v for k, v in hf.items() if k in keys})
        tc = hf.get("text_config", hf.get("text", hf))
        vc = hf.get("vision_config", hf.get("vision", {}))
        text = LlamaConfig.from_hf_config(dict(tc, architectures=["Qwen2ForCausalLM"]))
        text.attention_bias = True   # qwen2 qkv bias
        rs = tc.get("rope_scaling") or tc.get("rope_parameters") or {}
        return cls(
            text=text,
            vision=VisionConfig(
                embed_dim=vc.get("embed_dim", 1280),
                depth=vc.get("depth", 32),
                num_heads=vc.get("num_heads", 16),
                mlp_ratio=vc.get("mlp_ratio", 4.0),
                patch_size=vc.get("patch_size", 14),
                temporal_patch_size=vc.get("temporal_patch_size", 2),
                spatial_merge_size=vc.get("spatial_merge_size", 2),
                in_channels=vc.get("in_channels", 3),
                hidden_size=tc.get("hidden_size", 3584),
            ),
            mrope_section=tuple(rs.get("mrope_section", (16, 24, 24))),
            image_token_id=hf.get("image_token_id", 151655),
            vision_start_token_id=hf.get("vision_start_token_id", 151652),
        )


# --------------------------------------------------------------- vision tower
def vision_block_positions(grid_thw: torch.Tensor, merge: int) -> torch.Tensor:
    """(h, w) per patch token, block-major over merge x merge blocks -> [N, 2]."""
    out = []
    for t, h, w in grid_thw.tolist():
        hp, wp = torch.meshgrid(torch.arange(h), torch.arange(w), indexing="ij")
        shape = (h // merge, merge, w // merge, merge)
        hp = hp.reshape(shape).transpose(1, 2).flatten()
        wp = wp.reshape(shape).transpose(1, 2).flatten()
        out.append(torch.stack([hp, wp], dim=-1).repeat(t, 1))
    return torch.cat(out, dim=0)


class VisionAttention(nn.Module):
    def __init__(self, cfg: VisionConfig):
        super().__init__()
        self.num_heads = cfg.num_heads
        self.head_dim = cfg.embed_dim // cfg.num_heads
        self.separate = cfg.qkv_separate
        if self.separate:       # omni naming: attn.q / attn.k / attn.v
            self.q = nn.Linear(cfg.embed_dim, cfg.embed_dim, bias=True)
            self.k = nn.Linear(cfg.embed_dim, cfg.embed_dim, bias=True)
            self.v = nn.Linear(cfg.embed_dim, cfg.embed_dim, bias=True)
        else:
            self.qkv = nn.Linear(cfg.embed_dim, cfg.embed_dim * 3, bias=True)
        self.proj = nn.Linear(cfg.embed_dim, cfg.embed_dim)

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                cu_seqlens: torch.Tensor) -> torch.Tensor:
        N = x.shape[0]
        if self.separate:
            q = self.q(x).reshape(N, self.num_heads, -1)
            k = self.k(x).reshape(N, self.num_heads, -1)
            v = self.v(x).reshape(N, self.num_heads, -1)
        else:
            q, k, v = self.qkv(x).reshape(N, 3, self.num_heads, -1) \
                .permute(1, 0, 2, 3).unbind(0)
        c, s = cos[:, None, :].float(), sin[:, None, :].float()

        def rot(t):
            tf = t.float()
            t1, t2 = tf.chunk(2, dim=-1)
            rh = torch.cat([-t2, t1], dim=-1)
            return (tf * c + rh * s).to(t.dtype)

        q, k = rot(q), rot(k)
        # per-image full (non-causal) attention
        outs = []
        for a, b in zip(cu_seqlens[:-1].tolist(), cu_seqlens[1:].tolist()):
            o = F.scaled_dot_product_attention(
                q[a:b].transpose(0, 1)[None], k[a:b].transpose(0, 1)[None],
                v[a:b].transpose(0, 1)[None])
            outs.append(o[0].transpose(0, 1))
        return self.proj(torch.cat(outs, dim=0).reshape(N, -1))


class _VisionRMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        xf = x.float()
        inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + self.eps)
        return (xf * inv).to(x.dtype) * self.weight


class VisionBlock(nn.Module):
    def __init__(self, cfg: VisionConfig):
        super().__init__()
        self.v25 = cfg.variant == "v2_5"
        norm_cls = _VisionRMSNorm if self.v25 else (lambda d: nn.LayerNorm(d, eps=1e-6))
        self.norm1 = norm_cls(cfg.embed_dim)
        self.norm2 = norm_cls(cfg.embed_dim)
        self.attn = VisionAttention(cfg)
        if self.v25:   # swiglu with biases
            hidden = cfg.intermediate_size or int(cfg.embed_dim * cfg.mlp_ratio)
            self.mlp = nn.Module()
            self.mlp.gate_proj = nn.Linear(cfg.embed_dim, hidden, bias=True)
            self.mlp.up_proj = nn.Linear(cfg.embed_dim, hidden, bias=True)
            self.mlp.down_proj = nn.Linear(hidden, cfg.embed_dim, bias=True)
        else:
            hidden = int(cfg.embed_dim * cfg.mlp_ratio)
            self.mlp = nn.Sequential()
            self.mlp.fc1 = nn.Linear(cfg.embed_dim, hidden)
            self.mlp.fc2 = nn.Linear(hidden, cfg.embed_dim)

    def forward(self, x, cos, sin, cu):
        x = x + self.attn(self.norm1(x), cos, sin, cu)
        h = self.norm2(x)
        if self.v25:
            g = self.mlp.gate_proj(h)
            return x + self.mlp.down_proj(torch.nn.functional.silu(g) * self.mlp.up_proj(h))
        h = self.mlp.fc1(h)
        h = h * torch.sigmoid(1.702 * h)          # quick-gelu
        return x + self.mlp.fc2(h)


class PatchMerger(nn.Module):
    def __init__(self, cfg: VisionConfig):
        super().__init__()
        self.merge = cfg.spatial_merge_size
        dim = cfg.embed_dim * self.merge ** 2
        self.ln_q = (_VisionRMSNorm(cfg.embed_dim) if cfg.variant == "v2_5"
                     else nn.LayerNorm(cfg.embed_dim, eps=1e-6))
        self.mlp = nn.Sequential(nn.Linear(dim, dim), nn.GELU(),
                                 nn.Linear(dim, cfg.hidden_size))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.mlp(self.ln_q(x).reshape(-1, self.mlp[0].in_features))


class VisionTransformer(nn.Module):
    def __init__(self, cfg: VisionConfig):
        super().__init__()
        self.cfg = cfg
        self.spatial_merge_size = cfg.spatial_merge_size
        self.patch_embed = nn.Module()
        self.patch_embed.proj = nn.Conv3d(
            cfg.in_channels, cfg.embed_dim,
            kernel_size=[cfg.temporal_patch_size, cfg.patch_size, cfg.patch_size],
            stride=[cfg.temporal_patch_size, cfg.patch_size, cfg.patch_size],
            bias=False)
        self.blocks = nn.ModuleList(VisionBlock(cfg) for _ in range(cfg.depth))
        self.merger = PatchMerger(cfg)
        head_dim = cfg.embed_dim // cfg.num_heads
        inv = 1.0 / (10000.0 ** (torch.arange(0, head_dim // 2, 2,
                                              dtype=torch.float32) / (head_dim // 2)))
        self.register_buffer("rot_inv_freq", inv, persistent=False)

    def forward(self, pixel_values: torch.Tensor, grid_thw: torch.Tensor) -> torch.Tensor:
        """pixel_values [N, C * tp * p * p] flattened patches; -> merged
        embeddings [N / merge^2, text_hidden]."""
        c = self.cfg
        x = pixel_values.view(-1, c.in_channels, c.temporal_patch_size,
                              c.patch_size, c.patch_size)
        x = self.patch_embed.proj(x.to(self.patch_embed.proj.weight.dtype)) \
            .view(-1, c.embed_dim)
        pos = vision_block_positions(grid_thw, c.spatial_merge_size).to(x.device)
        freqs = (pos.unsqueeze(-1).float() * self.rot_inv_freq).flatten(1)
        emb = torch.cat([freqs, freqs], dim=-1)
        cos, sin = emb.cos(), emb.sin()
        cu = torch.cat([torch.zeros(1, dtype=torch.long,
                                    device=grid_thw.device),
                        grid_thw.prod(-1).cumsum(0)]).to(x.device)
        if c.variant == "v2_5":
            # window reorder (qwen2.5): merge-unit groups permuted so each
            # attention window is contiguous; inverted after the merger
            widx, cu_win = vision_window_index(grid_thw, c.spatial_merge_size,
                                               c.window_size, c.patch_size)
            widx = widx.to(x.device)
            mu = c.spatial_merge_size ** 2
            N = x.shape[0]
            x = x.reshape(N // mu, mu, -1)[widx].reshape(N, -1)
            cos = cos.reshape(N // mu, mu, -1)[widx].reshape(N, -1)
            sin = sin.reshape(N // mu, mu, -1)[widx].reshape(N, -1)
            cu_win = cu_win.to(x.device)
            for i, blk in enumerate(self.blocks):
                use_full = i in c.fullatt_block_indexes
                x = blk(x, cos, sin, cu if use_full else cu_win)
            out = self.merger(x)
            return out[torch.argsort(widx)]
        for blk in self.blocks:
            x = blk(x, cos, sin, cu)
        return self.merger(x)


def vision_window_index(grid_thw: torch.Tensor, merge: int, window: int,
                        patch: int) -> tuple[torch.Tensor, torch.Tensor]:
    """Window partition order over merge-unit groups (qwen2.5-vl): -> (index
    [N/merge^2], cu_window_seqlens in PATCH tokens)."""
    vw = window // merge // patch
    mu = merge ** 2
    idx_parts, cu = [], [0]
    base = 0
    for t, h, w in grid_thw.tolist():
        gh, gw = h // merge, w // merge
        index = torch.arange(t * gh * gw).reshape(t, gh, gw)
        pad_h, pad_w = (-gh) % vw, (-gw) % vw
        nh, nw = (gh + pad_h) // vw, (gw + pad_w) // vw
        padded = torch.nn.functional.pad(index, (0, pad_w, 0, pad_h),
                                         value=-100)
        padded = padded.reshape(t, nh, vw, nw, vw).permute(0, 1, 3, 2, 4) \
            .reshape(t, nh * nw, vw, vw)
        seqlens = (padded != -100).sum([2, 3]).reshape(-1)
        flat = padded.reshape(-1)
        idx_parts.append(flat[flat != -100] + base)
        cu.extend((seqlens.cumsum(0) * mu + cu[-1]).tolist())
        base += t * gh * gw
    cu_t = torch.unique_consecutive(torch.tensor(cu, dtype=torch.long))
    return torch.cat(idx_parts), cu_t


# ----------------------------------------------------------------- text model
def merge_mrope_tables(cos3: torch.Tensor, sin3: torch.Tensor,
                       sections: tuple) -> tuple[torch.Tensor, torch.Tensor]:
    """cos3/sin3 [3, B, S, D] -> [B, S, D] picking section i%3 channel-wise
    (HF apply_multimodal_rotary_pos_emb's channel interleave)."""
    secs = list(sections) * 2
    cos = torch.cat([m[i % 3] for i, m in enumerate(cos3.split(secs, dim=-1))], dim=-1)
    sin = torch.cat([m[i % 3] for i, m in enumerate(sin3.split(secs, dim=-1))], dim=-1)
    return cos, sin


class Qwen2VLForConditionalGeneration(nn.Module):
    hf_architectures = ("Qwen2VLForConditionalGeneration",)
    config_class = Qwen2VLConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> Qwen2VLConfig:
        return Qwen2VLConfig.from_hf_config(hf_cfg)

    def __init__(self, config: Qwen2VLConfig | dict,
                 backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = Qwen2VLConfig(**config)
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend),
            device_type, head_dim=config.text.head_dim,
        )
        self.config = config
        self.backend = backend
        tc = config.text
        inner = nn.Module()
        inner.visual = VisionTransformer(config.vision)
        lm = nn.Module()
        lm.embed_tokens = nn.Embedding(tc.vocab_size, tc.hidden_size)
        lm.layers = nn.ModuleList(LlamaDecoderLayer(tc, backend)
                                  for _ in range(tc.num_hidden_layers))
        lm.norm = RMSNorm(tc.hidden_size, tc.rms_norm_eps, backend.rms_norm)
        inv = 1.0 / (tc.rope_theta ** (torch.arange(0, tc.head_dim, 2,
                                                    dtype=torch.float32) / tc.head_dim))
        lm.register_buffer("rope_inv_freq", inv, persistent=False)
        inner.language_model = lm
        self.model = inner
        self.lm_head = nn.Linear(tc.hidden_size, tc.vocab_size, bias=False)
        if tc.tie_word_embeddings:
            self.lm_head.weight = lm.embed_tokens.weight
        self.loss_fn = None

    # -- m-rope position assignment (HF get_rope_index semantics) ------------
    def get_rope_index(self, input_ids: torch.Tensor,
                       image_grid_thw: torch.Tensor | None) -> torch.Tensor:
        B, S = input_ids.shape
        pos = torch.zeros(3, B, S, dtype=torch.long, device=input_ids.device)
        merge = self.config.vision.spatial_merge_size
        img_iter = iter(image_grid_thw.tolist()) if image_grid_thw is not None else iter([])
        for b in range(B):
            is_img = (input_ids[b] == self.config.image_token_id)
            cur = 0
            i = 0
            while i < S:
                j = i
                while j < S and bool(is_img[j]) == bool(is_img[i]):
                    j += 1
                n = j - i
                if not is_img[i]:
                    pos[:, b, i:j] = torch.arange(cur, cur + n,
                                                  device=input_ids.device)
                    cur += n
                else:
                    t, h, w = next(img_iter)
                    gt, gh, gw = t, h // merge, w // merge
                    tg = torch.arange(gt).repeat_interleave(gh * gw)
                    hg = torch.arange(gh).repeat_interleave(gw).repeat(gt)
                    wg = torch.arange(gw).repeat(gh * gt)
                    grid = torch.stack([tg, hg, wg]).to(input_ids.device) + cur
                    pos[:, b, i:j] = grid
                    cur = int(grid.max()) + 1
                i = j
        return pos

    def forward(self, input_ids: torch.Tensor,
                pixel_values: torch.Tensor | None = None,
                image_grid_thw: torch.Tensor | None = None,
                labels: torch.Tensor | None = None,
                position_ids: torch.Tensor | None = None, **_: Any) -> torch.Tensor:
        lm = self.model.language_model
        x = lm.embed_tokens(input_ids)
        if pixel_values is not None:
            img_embeds = self.model.visual(pixel_values, image_grid_thw)
            mask = input_ids == self.config.image_token_id
            x = x.clone()
            x[mask] = img_embeds.to(x.dtype)
        if position_ids is None or position_ids.dim() != 3:
            position_ids = self.get_rope_index(input_ids, image_grid_thw)
        freqs = position_ids[..., None].float() * lm.rope_inv_freq  # [3,B,S,D/2]
        emb = torch.cat([freqs, freqs], dim=-1)
        cos, sin = merge_mrope_tables(emb.cos(), emb.sin(),
                                      self.config.mrope_section)
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
            inv = 1.0 / (tc.rope_theta ** (torch.arange(
                0, tc.head_dim, 2, dtype=torch.float32) / tc.head_dim))
            self.model.language_model.rope_inv_freq.copy_(
                inv.to(self.model.language_model.rope_inv_freq.device))
            vt = self.model.visual
            hd = self.config.vision.embed_dim // self.config.vision.num_heads
            vinv = 1.0 / (10000.0 ** (torch.arange(0, hd // 2, 2,
                                                   dtype=torch.float32) / (hd // 2)))
            vt.rot_inv_freq.copy_(vinv.to(vt.rot_inv_freq.device))
        std = 0.02
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding, nn.Conv3d)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, (RMSNorm, nn.LayerNorm)):
                nn.init.ones_(mod.weight)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
        if self.config.text.tie_word_embeddings:
            self.lm_head.weight = self.model.language_model.embed_tokens.weight


class Qwen2_5_VLForConditionalGeneration(Qwen2VLForConditionalGeneration):
    """Qwen2.5-VL: the same m-rope text decoder with the v2_5 vision tower
    (RMSNorm, biased SwiGLU MLP, windowed attention with token reorder)."""

    hf_architectures = ("Qwen2_5_VLForConditionalGeneration",)

    @staticmethod
    def config_from_hf(hf_cfg) -> Qwen2VLConfig:
        if hasattr(hf_cfg, "to_dict"):
            hf_cfg = hf_cfg.to_dict()
        if "text" in hf_cfg and "vision" in hf_cfg:   # our exported dump
            return Qwen2VLConfig.from_hf_config(hf_cfg)
        tc, vc = hf_cfg.get("text_config", hf_cfg), hf_cfg.get("vision_config", {})
        text = LlamaConfig.from_hf_config(dict(tc, architectures=["Qwen2ForCausalLM"]))
        text.attention_bias = True
        rs = tc.get("rope_scaling") or tc.get("rope_parameters") or {}
        return Qwen2VLConfig(
            text=text,
            vision=VisionConfig(
                variant="v2_5",
                embed_dim=vc.get("hidden_size", 1280),     # 2.5 names it hidden_size
                depth=vc.get("depth", 32),
                num_heads=vc.get("num_heads", 16),
                intermediate_size=vc.get("intermediate_size"),
                patch_size=vc.get("patch_size", 14),
                temporal_patch_size=vc.get("temporal_patch_size", 2),
                spatial_merge_size=vc.get("spatial_merge_size", 2),
                in_channels=vc.get("in_channels", 3),
                hidden_size=vc.get("out_hidden_size", tc.get("hidden_size", 3584)),
                window_size=vc.get("window_size", 112),
                fullatt_block_indexes=tuple(vc.get("fullatt_block_indexes",
                                                   (7, 15, 23, 31))),
            ),
            mrope_section=tuple(rs.get("mrope_section", (16, 24, 24))),
            image_token_id=hf_cfg.get("image_token_id", 151655),
            vision_start_token_id=hf_cfg.get("vision_start_token_id", 151652),
        )
