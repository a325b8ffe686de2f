"""Qwen2.5-Omni thinker (audio + vision + text with TMRoPE), MI355X-native.

Reference behavior: nemo_automodel's omni family (components/models/
qwen2_5_omni/, recipes/multimodal). Implemented directly against the public
Qwen2.5-Omni thinker architecture:

  * audio tower: mel features chunked into ``n_window*2``-frame windows,
    conv1(gelu, pad-masked) + stride-2 conv2(gelu) per chunk, sinusoidal
    positions restarting per chunk, packed per-chunk bidirectional attention
    (cu_seqlens windows; q/v/out biased, k bias-free), per-SAMPLE stride-2
    average pooling over the packed layout, ln_post, and a linear ``proj``
    straight into the text width;
  * vision tower: the Qwen2.5-VL windowed ViT (RMSNorm, biased SwiGLU,
    window token reorder) with SEPARATE q/k/v projections (``attn.q/k/v``);
  * text: qwen2 stack (biased qkv) under multimodal 3-D rope, cos/sin merged
    channel-wise by ``mrope_section``;
  * TMRoPE ``get_rope_index``: text runs count 1-D; audio tokens advance the
    temporal axis only; images advance (t, h, w) with t stepping
    ``position_id_per_seconds`` per frame; videos step
    ``second_per_grid * position_id_per_seconds``; with
    ``use_audio_in_video`` the video and audio positions are interleaved in
    ``seconds_per_chunk`` time chunks sharing the same temporal origin.

State-dict keys match HF Qwen2_5OmniThinkerForConditionalGeneration
(parity-tested: text-only, +audio, +image paths).
"""

from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.llama.model import LlamaConfig, LlamaDecoderLayer
from automodel_amd.models.qwen2_vl.model import (
    VisionConfig,
    VisionTransformer,
    merge_mrope_tables,
)
from automodel_amd.ops.rms_norm import RMSNorm


@dataclass
class OmniAudioConfig:
    d_model: int = 1280
    encoder_layers: int = 32
    encoder_attention_heads: int = 20
    encoder_ffn_dim: int = 5120
    num_mel_bins: int = 128
    max_source_positions: int = 1500
    n_window: int = 100
    output_dim: int = 3584
    activation_function: str = "gelu"


@dataclass
class Qwen2_5OmniThinkerConfig:
    text: dict = field(default_factory=dict)
    audio: OmniAudioConfig = field(default_factory=OmniAudioConfig)
    vision: VisionConfig = field(default_factory=VisionConfig)
    mrope_section: tuple = (16, 24, 24)
    audio_token_id: int = 151646
    image_token_id: int = 151655
    video_token_id: int = 151656
    vision_start_token_id: int = 151652
    audio_start_token_id: int = 151647
    position_id_per_seconds: int = 25
    seconds_per_chunk: int = 2
    initializer_range: float = 0.02

    def __post_init__(self):
        if isinstance(self.audio, dict):
            self.audio = OmniAudioConfig(**self.audio)
        if isinstance(self.vision, dict):
            self.vision = VisionConfig(**self.vision)
        self.mrope_section = tuple(self.mrope_section)

    @property
    def vocab_size(self):
        return self.text_cfg.vocab_size

    @property
    def num_hidden_layers(self):
        return self.text_cfg.num_hidden_layers

    @property
    def text_cfg(self) -> LlamaConfig:
        if not isinstance(self.text, LlamaConfig):
            self.text = (LlamaConfig(**self.text) if "hidden_size" in self.text
                         else LlamaConfig.from_hf_config(dict(self.text)))
        return self.text

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Qwen2_5OmniThinkerConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        tc = dict(hf.get("text_config", {}))
        text = LlamaConfig.from_hf_config(dict(tc, architectures=["Qwen2ForCausalLM"]))
        text.attention_bias = True
        a = hf.get("audio_config", {})
        vc = hf.get("vision_config", {})
        rs = (tc.get("rope_scaling") or tc.get("rope_parameters") or {})
        return cls(
            text=text,
            audio=OmniAudioConfig(
                d_model=a.get("d_model", 1280),
                encoder_layers=a.get("encoder_layers", 32),
                encoder_attention_heads=a.get("encoder_attention_heads", 20),
                encoder_ffn_dim=a.get("encoder_ffn_dim", 5120),
                num_mel_bins=a.get("num_mel_bins", 128),
                max_source_positions=a.get("max_source_positions", 1500),
                n_window=a.get("n_window", 100),
                output_dim=a.get("output_dim", tc.get("hidden_size", 3584)),
                activation_function=a.get("activation_function", "gelu"),
            ),
            vision=VisionConfig(
                variant="v2_5",
                qkv_separate=True,
                embed_dim=vc.get("hidden_size", 1280),
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
            audio_token_id=hf.get("audio_token_id", hf.get("audio_token_index", 151646)),
            image_token_id=hf.get("image_token_id", hf.get("image_token_index", 151655)),
            video_token_id=hf.get("video_token_id", hf.get("video_token_index", 151656)),
            vision_start_token_id=hf.get("vision_start_token_id", 151652),
            audio_start_token_id=hf.get("audio_start_token_id", 151647),
            position_id_per_seconds=hf.get("position_id_per_seconds", 25),
            seconds_per_chunk=hf.get("seconds_per_chunk", 2),
            initializer_range=hf.get("initializer_range", 0.02),
        )


# --------------------------------------------------------------- audio tower
class OmniAudioEncoderLayer(nn.Module):
    """Pre-LN bidirectional layer; attention is windowed by cu_seqlens."""

    def __init__(self, cfg: OmniAudioConfig):
        super().__init__()
        D = cfg.d_model
        self.n_heads = cfg.encoder_attention_heads
        attn = nn.Module()
        attn.q_proj = nn.Linear(D, D, bias=True)
        attn.k_proj = nn.Linear(D, D, bias=False)
        attn.v_proj = nn.Linear(D, D, bias=True)
        attn.out_proj = nn.Linear(D, D, bias=True)
        self.self_attn = attn
        self.self_attn_layer_norm = nn.LayerNorm(D)
        self.fc1 = nn.Linear(D, cfg.encoder_ffn_dim)
        self.fc2 = nn.Linear(cfg.encoder_ffn_dim, D)
        self.final_layer_norm = nn.LayerNorm(D)
        self.act = (F.gelu if cfg.activation_function == "gelu"
                    else getattr(F, cfg.activation_function))

    def forward(self, x: torch.Tensor, cu: torch.Tensor) -> torch.Tensor:
        # x: [N, D] packed chunk tokens
        a = self.self_attn
        h = self.self_attn_layer_norm(x)
        N = h.shape[0]
        q = a.q_proj(h).reshape(N, self.n_heads, -1)
        k = a.k_proj(h).reshape(N, self.n_heads, -1)
        v = a.v_proj(h).reshape(N, self.n_heads, -1)
        outs = []
        for s, e in zip(cu[:-1].tolist(), cu[1:].tolist()):
            o = F.scaled_dot_product_attention(
                q[s:e].transpose(0, 1)[None], k[s:e].transpose(0, 1)[None],
                v[s:e].transpose(0, 1)[None])
            outs.append(o[0].transpose(0, 1))
        x = x + a.out_proj(torch.cat(outs, dim=0).reshape(N, -1))
        return x + self.fc2(self.act(self.fc1(self.final_layer_norm(x))))


def _sinusoids(length: int, channels: int, max_timescale: float = 10000.0):
    inc = math.log(max_timescale) / (channels // 2 - 1)
    inv = torch.exp(-inc * torch.arange(channels // 2).float())
    t = torch.arange(length).float()[:, None] * inv[None, :]
    return torch.cat([t.sin(), t.cos()], dim=1)


class OmniAudioEncoder(nn.Module):
    """Windowed Whisper-style encoder: chunk -> conv -> packed attention ->
    per-sample stride-2 pool -> ln_post -> proj (into the text width)."""

    def __init__(self, cfg: OmniAudioConfig):
        super().__init__()
        D = cfg.d_model
        self.cfg = cfg
        self.conv1 = nn.Conv1d(cfg.num_mel_bins, D, kernel_size=3, padding=1)
        self.conv2 = nn.Conv1d(D, D, kernel_size=3, stride=2, padding=1)
        self.register_buffer("positional_embedding",
                             _sinusoids(cfg.max_source_positions, D),
                             persistent=False)
        # present in HF checkpoints (used by the legacy processing path)
        self.audio_bos_eos_token = nn.Embedding(2, cfg.output_dim)
        self.layers = nn.ModuleList(OmniAudioEncoderLayer(cfg)
                                    for _ in range(cfg.encoder_layers))
        self.ln_post = nn.LayerNorm(D)
        self.proj = nn.Linear(D, cfg.output_dim)

    @staticmethod
    def output_lengths(feature_lens: torch.Tensor) -> torch.Tensor:
        """mel frames -> LLM audio tokens: conv2 stride 2 then pool stride 2."""
        after = (feature_lens - 1) // 2 + 1
        return (after - 2) // 2 + 1

    def forward(self, input_features: torch.Tensor,
                feature_lens: torch.Tensor) -> torch.Tensor:
        """input_features: [mel, total_frames] packed across audios;
        feature_lens: [num_audios]. Returns [total_audio_tokens, output_dim]."""
        nw2 = self.cfg.n_window * 2
        dev = input_features.device
        # ---- chunk to <=nw2-frame windows, right-pad to the longest chunk
        chunk_num = torch.ceil(feature_lens / nw2).long()
        chunk_lengths = torch.full((int(chunk_num.sum()),), nw2,
                                   dtype=torch.long, device=dev)
        tail = F.pad(chunk_num, (1, 0), value=-1).cumsum(0)[1:]
        chunk_lengths[tail] = feature_lens % nw2
        chunk_lengths = torch.where(chunk_lengths == 0, nw2, chunk_lengths)
        chunks = input_features.T.split(chunk_lengths.tolist(), dim=0)
        padded = nn.utils.rnn.pad_sequence(chunks, batch_first=True).transpose(1, 2)
        pad_mask = (torch.arange(padded.shape[2], device=dev)
                    < chunk_lengths[:, None]).unsqueeze(1)
        # ---- conv stack (conv1 output masked at padded frames)
        x = F.gelu(self.conv1(padded.to(self.conv1.weight.dtype))) * pad_mask
        x = F.gelu(self.conv2(x)).transpose(1, 2)          # [C, L2, D]
        x = x + self.positional_embedding[: x.shape[1]].to(x.dtype)
        # ---- pack valid post-conv positions
        after1 = (chunk_lengths - 1) // 2 + 1
        maxlen = int(after1.max())
        valid = (torch.arange(maxlen, device=dev) < after1[:, None]) \
            .flatten().nonzero().squeeze(-1)
        h = x.reshape(-1, x.shape[-1])[valid]
        cu = F.pad(after1.cumsum(0), (1, 0), value=0)
        for layer in self.layers:
            h = layer(h, cu)
        # ---- per-sample stride-2 average pool over the packed layout
        after_s = (feature_lens - 1) // 2 + 1
        num_pooled = (after_s - 2) // 2 + 1
        offs = F.pad(after_s[:-1].cumsum(0), (1, 0), value=0)
        pair_offs = torch.repeat_interleave(offs, num_pooled)
        local = torch.arange(int(num_pooled.sum()), device=dev)
        local = local - torch.repeat_interleave(
            F.pad(num_pooled[:-1].cumsum(0), (1, 0), value=0), num_pooled)
        pool_idx = pair_offs + local * 2
        h = (h[pool_idx] + h[pool_idx + 1]) / 2
        return self.proj(self.ln_post(h))


# ------------------------------------------------------------------ TMRoPE
def _text_pos(n: int, start: int) -> torch.Tensor:
    return torch.arange(start, start + n).view(1, -1).expand(3, -1)


def _vision_pos(start: int, t_index: list[int], gh: int, gw: int) -> torch.Tensor:
    nt = len(t_index)
    hh = torch.arange(gh).view(1, -1, 1).expand(nt, -1, gw).flatten()
    ww = torch.arange(gw).view(1, 1, -1).expand(nt, gh, -1).flatten()
    tt = torch.tensor(t_index).view(-1, 1).expand(-1, gh * gw).flatten().long()
    return torch.stack([tt, hh, ww]) + start


def _chunk_spans(t_axis: torch.Tensor, per_chunk: int, origin: int):
    """(start, end) index spans grouping a monotone temporal axis into
    per_chunk-sized time windows (TMRoPE audio/video interleave)."""
    spans, s, chunk = [], 0, 1
    for i in range(len(t_axis)):
        if int(t_axis[i]) - origin >= chunk * per_chunk:
            spans.append((s, i))
            s = i
            chunk += 1
    spans.append((s, len(t_axis)))
    return spans


class Qwen2_5OmniThinkerForConditionalGeneration(nn.Module):
    hf_architectures = ("Qwen2_5OmniThinkerForConditionalGeneration",)
    config_class = Qwen2_5OmniThinkerConfig

    @staticmethod
    def config_from_hf(hf_cfg) -> Qwen2_5OmniThinkerConfig:
        return Qwen2_5OmniThinkerConfig.from_hf_config(hf_cfg)

    def __init__(self, config: Qwen2_5OmniThinkerConfig | dict,
                 backend: BackendConfig | dict | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = Qwen2_5OmniThinkerConfig(**config)
        self.config = config
        tc = config.text_cfg
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
        backend = BackendConfig.resolve(
            backend if not isinstance(backend, dict) else BackendConfig(**backend),
            device_type, head_dim=tc.head_dim)
        self.backend = backend
        self.audio_tower = OmniAudioEncoder(config.audio)
        self.visual = VisionTransformer(config.vision)
        lm = nn.Module()
        lm.embed_tokens = nn.Embedding(tc.vocab_size, tc.hidden_size)
        lm.layers = nn.ModuleList(LlamaDecoderLayer(tc, backend)
                                  for _ in range(tc.num_hidden_layers))
        lm.norm = RMSNorm(tc.hidden_size, tc.rms_norm_eps, backend.rms_norm)
        inv = 1.0 / (tc.rope_theta ** (torch.arange(0, tc.head_dim, 2,
                                                    dtype=torch.float32) / tc.head_dim))
        lm.register_buffer("rope_inv_freq", inv, persistent=False)
        self.model = lm
        self.lm_head = nn.Linear(tc.hidden_size, tc.vocab_size, bias=False)
        if tc.tie_word_embeddings:
            self.lm_head.weight = lm.embed_tokens.weight
        self.loss_fn = None

    # ---- TMRoPE position assignment --------------------------------------
    def get_rope_index(self, input_ids: torch.Tensor,
                       image_grid_thw: torch.Tensor | None = None,
                       video_grid_thw: torch.Tensor | None = None,
                       audio_seqlens: torch.Tensor | None = None,
                       second_per_grids: torch.Tensor | None = None,
                       use_audio_in_video: bool = False) -> torch.Tensor:
        cfg = self.config
        B, S = input_ids.shape
        pps = cfg.position_id_per_seconds
        merge = cfg.vision.spatial_merge_size
        pos = torch.zeros(3, B, S, dtype=torch.long, device=input_ids.device)
        img_i = vid_i = aud_i = 0
        for b in range(B):
            toks = input_ids[b].tolist()
            parts: list[torch.Tensor] = []

            def nxt():
                return int(parts[-1].max()) + 1 if parts else 0

            st = 0
            while st < len(toks):
                try:
                    ed_a = toks.index(cfg.audio_token_id, st)
                except ValueError:
                    ed_a = len(toks) + 1
                try:
                    ed_i = toks.index(cfg.image_token_id, st)
                except ValueError:
                    ed_i = len(toks) + 1
                try:
                    ed_v = toks.index(cfg.video_token_id, st)
                except ValueError:
                    ed_v = len(toks) + 1
                ed = min(ed_a, ed_i, ed_v)
                if ed > len(toks):
                    parts.append(_text_pos(len(toks) - st, nxt()))
                    break
                if ed == ed_a and not (use_audio_in_video and ed == ed_v):
                    # [text][audio_bos][audio][audio_eos]
                    if ed - st - 1 > 0:
                        parts.append(_text_pos(ed - st - 1, nxt()))
                    parts.append(_text_pos(1, nxt()))            # bos
                    alen = int(OmniAudioEncoder.output_lengths(
                        audio_seqlens[aud_i]))
                    parts.append(_text_pos(alen, nxt()))
                    parts.append(_text_pos(1, nxt()))            # eos
                    st = ed + alen + 1
                    aud_i += 1
                elif ed == ed_i:
                    if ed - st - 1 > 0:
                        parts.append(_text_pos(ed - st - 1, nxt()))
                    parts.append(_text_pos(1, nxt()))            # vision bos
                    t, h, w = image_grid_thw[img_i].tolist()
                    t_idx = (torch.arange(t) * 1 * pps).long().tolist()
                    parts.append(_vision_pos(nxt(), t_idx, h // merge, w // merge))
                    parts.append(_text_pos(1, nxt()))            # vision eos
                    st = ed + t * (h // merge) * (w // merge) + 1
                    img_i += 1
                elif ed == ed_v and not use_audio_in_video:
                    if ed - st - 1 > 0:
                        parts.append(_text_pos(ed - st - 1, nxt()))
                    parts.append(_text_pos(1, nxt()))
                    t, h, w = video_grid_thw[vid_i].tolist()
                    spg = (float(second_per_grids[vid_i])
                           if second_per_grids is not None else 1.0)
                    t_idx = (torch.arange(t) * spg * pps).long().tolist()
                    parts.append(_vision_pos(nxt(), t_idx, h // merge, w // merge))
                    parts.append(_text_pos(1, nxt()))
                    st = ed + t * (h // merge) * (w // merge) + 1
                    vid_i += 1
                else:
                    # video with its audio track: [text][v_bos][a_bos]
                    # [chunked video x audio interleave][v_eos][a_eos]
                    if ed - st - 2 > 0:
                        parts.append(_text_pos(ed - st - 2, nxt()))
                    bos = _text_pos(1, nxt())
                    parts.append(bos)            # vision bos and audio bos
                    parts.append(bos.clone())    # share one position
                    origin = nxt()
                    alen = int(OmniAudioEncoder.output_lengths(
                        audio_seqlens[aud_i]))
                    a_pos = _text_pos(alen, origin)
                    t, h, w = video_grid_thw[vid_i].tolist()
                    spg = (float(second_per_grids[vid_i])
                           if second_per_grids is not None else 1.0)
                    t_idx = (torch.arange(t) * spg * pps).long().tolist()
                    v_pos = _vision_pos(origin, t_idx, h // merge, w // merge)
                    per_chunk = int(pps * cfg.seconds_per_chunk)
                    v_spans = _chunk_spans(v_pos[0], per_chunk, origin)
                    a_spans = _chunk_spans(a_pos[0], per_chunk, origin)
                    for j in range(max(len(v_spans), len(a_spans))):
                        if j < len(v_spans):
                            parts.append(v_pos[:, v_spans[j][0]:v_spans[j][1]])
                        if j < len(a_spans):
                            parts.append(a_pos[:, a_spans[j][0]:a_spans[j][1]])
                    eos = _text_pos(1, nxt())
                    parts.append(eos)
                    parts.append(eos.clone())
                    vlen = t * (h // merge) * (w // merge)
                    st = ed + vlen + alen + 2
                    aud_i += 1
                    vid_i += 1
            pos[:, b] = torch.cat(parts, dim=1).to(input_ids.device)
        return pos

    # ---- forward ----------------------------------------------------------
    def forward(self, input_ids: torch.Tensor,
                input_features: torch.Tensor | None = None,
                feature_attention_mask: torch.Tensor | None = None,
                pixel_values: torch.Tensor | None = None,
                image_grid_thw: torch.Tensor | None = None,
                pixel_values_videos: torch.Tensor | None = None,
                video_grid_thw: torch.Tensor | None = None,
                video_second_per_grid: torch.Tensor | None = None,
                use_audio_in_video: bool = False,
                labels: torch.Tensor | None = None,
                position_ids: torch.Tensor | None = None, **_: Any):
        cfg = self.config
        x = self.model.embed_tokens(input_ids)
        audio_lens = None
        if input_features is not None:
            if feature_attention_mask is not None:
                audio_lens = feature_attention_mask.sum(-1)
                packed = input_features.permute(0, 2, 1)[
                    feature_attention_mask.bool()].T
            else:
                audio_lens = torch.tensor([input_features.shape[-1]] *
                                          input_features.shape[0],
                                          device=input_ids.device)
                packed = input_features.permute(1, 0, 2).reshape(
                    input_features.shape[1], -1)
            audio_embeds = self.audio_tower(packed.to(x.dtype), audio_lens)
            mask = input_ids == cfg.audio_token_id
            x = x.clone()
            x[mask] = audio_embeds.to(x.dtype)
        if pixel_values is not None:
            img = self.visual(pixel_values, image_grid_thw)
            mask = input_ids == cfg.image_token_id
            x = x.clone()
            x[mask] = img.to(x.dtype)
        if pixel_values_videos is not None:
            vid = self.visual(pixel_values_videos, video_grid_thw)
            mask = input_ids == cfg.video_token_id
            x = x.clone()
            x[mask] = vid.to(x.dtype)
        if position_ids is None or position_ids.dim() != 3:
            position_ids = self.get_rope_index(
                input_ids, image_grid_thw, video_grid_thw, audio_lens,
                video_second_per_grid, use_audio_in_video)
        freqs = position_ids[..., None].float() * self.model.rope_inv_freq
        emb = torch.cat([freqs, freqs], dim=-1)
        cos, sin = merge_mrope_tables(emb.cos(), emb.sin(), cfg.mrope_section)
        for layer in self.model.layers:
            x = layer(x, cos, sin)
        hidden = self.model.norm(x)
        if labels is not None and self.loss_fn is not None:
            return self.loss_fn(hidden, self.lm_head.weight, labels)
        logits = self.lm_head(hidden)
        if labels is not None:
            return F.cross_entropy(
                logits.reshape(-1, logits.shape[-1]).float(), labels.reshape(-1),
                ignore_index=-100, reduction="sum")
        return logits

    def freeze_towers(self) -> None:
        for p in (*self.audio_tower.parameters(), *self.visual.parameters()):
            p.requires_grad_(False)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            tc = self.config.text_cfg
            inv = 1.0 / (tc.rope_theta ** (torch.arange(
                0, tc.head_dim, 2, dtype=torch.float32) / tc.head_dim))
            self.model.rope_inv_freq.copy_(inv.to(self.model.rope_inv_freq.device))
            a = self.config.audio
            self.audio_tower.positional_embedding.copy_(
                _sinusoids(a.max_source_positions, a.d_model).to(
                    self.audio_tower.positional_embedding.device))
            vt = self.visual
            hd = self.config.vision.embed_dim // self.config.vision.num_heads
            vinv = 1.0 / (10000.0 ** (torch.arange(0, hd // 2, 2,
                                                   dtype=torch.float32) / (hd // 2)))
            vt.rot_inv_freq.copy_(vinv.to(vt.rot_inv_freq.device))
        std = self.config.initializer_range
        for mod in self.modules():
            if isinstance(mod, (nn.Linear, nn.Embedding, nn.Conv1d, nn.Conv3d)):
                nn.init.normal_(mod.weight, std=std)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
            elif isinstance(mod, (RMSNorm, nn.LayerNorm)):
                nn.init.ones_(mod.weight)
                if getattr(mod, "bias", None) is not None:
                    nn.init.zeros_(mod.bias)
        if self.config.text_cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
