"""Qwen3-Omni-MoE thinker (AuT audio + DeepStack ViT + MoE text), MI355X-native.

Reference behavior: nemo_automodel's omni family (components/models/
qwen3_omni_moe/, recipes/multimodal). Implemented directly against the
public Qwen3-Omni-MoE thinker architecture:

  * audio tower (AuT): mel features chunked into ``n_window*2``-frame
    windows, 3x stride-2 Conv2d downsampling over the (mel, time) image
    (8x in time), a ``conv_out`` linear folding (channels x remaining mel
    bins) into ``d_model``, sinusoidal positions restarting per chunk,
    packed bidirectional attention over MERGED inference windows
    (``n_window_infer`` raw frames per attention window, spanning several
    conv chunks), ln_post, then proj1 -> gelu -> proj2 into the text width;
  * vision tower: the Qwen3-VL DeepStack ViT (bilinear pos-table resample,
    2-axis rotary, per-image full attention, postshuffle deepstack
    mergers) with the omni merger naming (``merger_list``, ``ln_q``,
    ``mlp.{0,2}``);
  * text: Qwen3-MoE stack (qk-norm attention, softmax-topk-renorm routed
    experts) under interleaved 3-D MRoPE, deepstack features added to
    vision token positions after the first N layers;
  * ``get_rope_index``: FLOAT positions; text counts 1-D, audio advances
    the temporal axis, vision advances (t, h, w) with t stepping
    ``position_id_per_seconds`` per frame (x second_per_grid for video);
    with audio-in-video the two streams are merge-sorted by temporal
    position and the bos/eos pairs take two slots each.

State-dict keys match HF Qwen3OmniMoeThinkerForConditionalGeneration via
the fused-expert adapter (gate_up_proj split shared with Qwen3-VL-MoE).
"""

from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Any

import torch
import torch.nn as nn
import torch.nn.functional as F

from automodel_amd.models.qwen3_vl.model import (
    Qwen3VLConfig,
    Qwen3VLMoeStateDictAdapter,
    Qwen3VLTextConfig,
    Qwen3VLTextModel,
    Qwen3VLVisionConfig,
    Qwen3VLVisionModel,
)
from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.qwen2_5_omni.model import _sinusoids
from automodel_amd.ops.rms_norm import RMSNorm


@dataclass
class AuTAudioConfig:
    d_model: int = 1280
    encoder_layers: int = 32
    encoder_attention_heads: int = 20
    encoder_ffn_dim: int = 5120
    num_mel_bins: int = 128
    max_source_positions: int = 1500
    n_window: int = 50
    n_window_infer: int = 800
    conv_chunksize: int = 500
    downsample_hidden_size: int = 480
    output_dim: int = 3584
    activation_function: str = "gelu"


def aut_output_lengths(lens, n_window: int):
    """mel frames -> audio tokens: per full chunk 3x ceil-halving of
    ``n_window*2`` frames; the remainder chunk is halved on its own."""
    chunk = n_window * 2
    leave = lens % chunk
    f = (leave - 1) // 2 + 1
    f = (f - 1) // 2 + 1
    f = (f - 1) // 2 + 1
    return f + (lens // chunk) * ((((chunk - 1) // 2 + 1 - 1) // 2 + 1 - 1) // 2 + 1)


@dataclass
class Qwen3OmniMoeConfig:
    text: Qwen3VLTextConfig = field(default_factory=Qwen3VLTextConfig)
    vision: Qwen3VLVisionConfig = field(default_factory=Qwen3VLVisionConfig)
    audio: AuTAudioConfig = field(default_factory=AuTAudioConfig)
    audio_token_id: int = 151646
    image_token_id: int = 151655
    video_token_id: int = 151656
    vision_start_token_id: int = 151652
    audio_start_token_id: int = 151647
    position_id_per_seconds: int = 25
    initializer_range: float = 0.02

    def __post_init__(self):
        if isinstance(self.text, dict):
            self.text = Qwen3VLTextConfig(**self.text)
        if isinstance(self.vision, dict):
            self.vision = Qwen3VLVisionConfig(**self.vision)
        if isinstance(self.audio, dict):
            self.audio = AuTAudioConfig(**self.audio)

    @property
    def vocab_size(self):
        return self.text.vocab_size

    @property
    def num_hidden_layers(self):
        return self.text.num_hidden_layers

    @classmethod
    def from_hf_config(cls, hf: Any) -> "Qwen3OmniMoeConfig":
        if hasattr(hf, "to_dict"):
            hf = hf.to_dict()
        base = Qwen3VLConfig.from_hf_config(hf)
        a = hf.get("audio_config", {})
        return cls(
            text=base.text, vision=base.vision,
            audio=AuTAudioConfig(
                d_model=a.get("d_model", 1280),
                encoder_layers=a.get("encoder_layers", 32),
                encoder_attention_heads=a.get("encoder_attention_heads", 20),
                encoder_ffn_dim=a.get("encoder_ffn_dim", 5120),
                num_mel_bins=a.get("num_mel_bins", 128),
                max_source_positions=a.get("max_source_positions", 1500),
                n_window=a.get("n_window", 50),
                n_window_infer=a.get("n_window_infer", 800),
                conv_chunksize=a.get("conv_chunksize", 500),
                downsample_hidden_size=a.get("downsample_hidden_size", 480),
                output_dim=a.get("output_dim", 3584),
                activation_function=a.get("activation_function", "gelu"),
            ),
            audio_token_id=hf.get("audio_token_id", 151646),
            image_token_id=hf.get("image_token_id", 151655),
            video_token_id=hf.get("video_token_id", 151656),
            vision_start_token_id=hf.get("vision_start_token_id", 151652),
            audio_start_token_id=hf.get("audio_start_token_id", 151647),
            position_id_per_seconds=hf.get("position_id_per_seconds", 25),
            initializer_range=hf.get("initializer_range", 0.02),
        )


# --------------------------------------------------------------- audio (AuT)
class AuTEncoderLayer(nn.Module):
    """Pre-LN bidirectional layer over packed inference windows (all four
    attention projections biased, unlike the 2.5 encoder's k_proj)."""

    def __init__(self, cfg: AuTAudioConfig):
        super().__init__()
        D = cfg.d_model
        self.n_heads = cfg.encoder_attention_heads
        attn = nn.Module()
        attn.q_proj = nn.Linear(D, D, bias=True)
        attn.k_proj = nn.Linear(D, D, bias=True)
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


class AuTAudioEncoder(nn.Module):
    def __init__(self, cfg: AuTAudioConfig):
        super().__init__()
        D, C = cfg.d_model, cfg.downsample_hidden_size
        self.cfg = cfg
        self.register_buffer("positional_embedding",
                             _sinusoids(cfg.max_source_positions, D),
                             persistent=False)
        self.layers = nn.ModuleList(AuTEncoderLayer(cfg)
                                    for _ in range(cfg.encoder_layers))
        self.ln_post = nn.LayerNorm(D)
        self.conv2d1 = nn.Conv2d(1, C, 3, 2, padding=1)
        self.conv2d2 = nn.Conv2d(C, C, 3, 2, padding=1)
        self.conv2d3 = nn.Conv2d(C, C, 3, 2, padding=1)
        mel_down = (((cfg.num_mel_bins + 1) // 2 + 1) // 2 + 1) // 2
        self.conv_out = nn.Linear(C * mel_down, D, bias=False)
        self.proj1 = nn.Linear(D, D)
        self.proj2 = nn.Linear(D, cfg.output_dim)
        self.act = (F.gelu if cfg.activation_function == "gelu"
                    else getattr(F, cfg.activation_function))

    @staticmethod
    def _ceil_half(x):
        return (x - 1) // 2 + 1

    def forward(self, input_features: torch.Tensor,
                feature_lens: torch.Tensor) -> torch.Tensor:
        """input_features [mel, total_frames] packed; feature_lens
        [num_audios] -> [total_audio_tokens, output_dim]."""
        cfg = self.cfg
        nw2 = cfg.n_window * 2
        dev = input_features.device
        chunk_num = torch.ceil(feature_lens / nw2).long()
        chunk_lengths = torch.full((int(chunk_num.sum()),), nw2,
                                   dtype=torch.long, device=dev)
        tail = F.pad(chunk_num, (1, 0), value=-1).cumsum(0)[1:]
        chunk_lengths[tail] = feature_lens % nw2
        chunk_lengths = torch.where(chunk_lengths == 0, nw2, chunk_lengths)
        chunks = input_features.T.split(chunk_lengths.tolist(), dim=0)
        padded = nn.utils.rnn.pad_sequence(chunks, batch_first=True).transpose(1, 2)
        # ---- 3x stride-2 conv2d over the (1, mel, time) image, chunked
        embeds = []
        for part in padded.unsqueeze(1).to(self.conv2d1.weight.dtype) \
                          .split(cfg.conv_chunksize, dim=0):
            e = F.gelu(self.conv2d1(part))
            e = F.gelu(self.conv2d2(e))
            embeds.append(F.gelu(self.conv2d3(e)))
        e = torch.cat(embeds, dim=0)
        b, c, f, t = e.shape
        x = self.conv_out(e.permute(0, 3, 1, 2).reshape(b, t, c * f))
        x = x + self.positional_embedding[: x.shape[1]].to(x.dtype)
        # ---- pack valid post-conv positions (per-chunk lengths)
        after_chunk = aut_output_lengths(chunk_lengths, cfg.n_window)
        maxlen = int(after_chunk.max())
        valid = (torch.arange(maxlen, device=dev) < after_chunk[:, None]) \
            .flatten().nonzero().squeeze(-1)
        h = x.reshape(-1, x.shape[-1])[valid]
        # ---- attention windows merging n_window_infer frames of chunks
        after_sample = aut_output_lengths(feature_lens, cfg.n_window)
        ratio = cfg.n_window_infer // nw2
        win = maxlen * ratio
        cu_list = [0]
        for ln in after_sample.tolist():
            cu_list += [win] * (ln // win)
            if ln % win:
                cu_list.append(ln % win)
        cu = torch.tensor(cu_list, device=dev).cumsum(0)
        for layer in self.layers:
            h = layer(h, cu)
        return self.proj2(self.act(self.proj1(self.ln_post(h))))


# --------------------------------------------------------------- vision
class OmniVisionPatchMerger(nn.Module):
    """Qwen3-VL merger with the omni key names (ln_q + mlp ModuleList)."""

    def __init__(self, cfg: Qwen3VLVisionConfig, postshuffle_norm: bool):
        super().__init__()
        merged = cfg.hidden_size * cfg.spatial_merge_size ** 2
        self.merged = merged
        self.postshuffle = postshuffle_norm
        self.ln_q = nn.LayerNorm(merged if postshuffle_norm else cfg.hidden_size,
                                 eps=1e-6)
        self.mlp = nn.ModuleList([nn.Linear(merged, merged), nn.GELU(),
                                  nn.Linear(merged, cfg.out_hidden_size)])

    def forward(self, x):
        x = self.ln_q(x.view(-1, self.merged) if self.postshuffle else x)
        return self.mlp[2](self.mlp[1](self.mlp[0](x.view(-1, self.merged))))


class Qwen3OmniVisionModel(Qwen3VLVisionModel):
    def __init__(self, cfg: Qwen3VLVisionConfig):
        super().__init__(cfg)
        del self._modules["merger"], self._modules["deepstack_merger_list"]
        self.merger = OmniVisionPatchMerger(cfg, postshuffle_norm=False)
        self.merger_list = nn.ModuleList(
            OmniVisionPatchMerger(cfg, postshuffle_norm=True)
            for _ in cfg.deepstack_visual_indexes)

    @property
    def deepstack_merger_list(self):      # base forward reads this name
        return self.merger_list


# --------------------------------------------------------------- thinker
def _ftext(n: int, start: float) -> torch.Tensor:
    return (torch.arange(n).float().view(1, -1).expand(3, -1) + start)


def _fvision(start: float, t_index: torch.Tensor, gh: int, gw: int) -> torch.Tensor:
    nt = len(t_index)
    hh = torch.arange(gh).view(1, -1, 1).expand(nt, -1, gw).flatten().float()
    ww = torch.arange(gw).view(1, 1, -1).expand(nt, gh, -1).flatten().float()
    tt = t_index.view(-1, 1).expand(-1, gh * gw).flatten().float()
    return torch.stack([tt, hh, ww]) + start


class Qwen3OmniMoeThinkerForConditionalGeneration(nn.Module):
    hf_architectures = ("Qwen3OmniMoeThinkerForConditionalGeneration",)
    config_class = Qwen3OmniMoeConfig
    state_dict_adapter = Qwen3VLMoeStateDictAdapter

    @staticmethod
    def config_from_hf(hf_cfg) -> Qwen3OmniMoeConfig:
        return Qwen3OmniMoeConfig.from_hf_config(hf_cfg)

    def __init__(self, config: Qwen3OmniMoeConfig | dict, backend=None):
        super().__init__()
        cfg = (config if isinstance(config, Qwen3OmniMoeConfig)
               else Qwen3OmniMoeConfig(**dict(config)))
        self.config = cfg
        bk = BackendConfig.resolve(backend,
                                   "cuda" if torch.cuda.is_available() else "cpu",
                                   head_dim=cfg.text.head_dim)
        self.audio_tower = AuTAudioEncoder(cfg.audio)
        self.visual = Qwen3OmniVisionModel(cfg.vision)
        self.model = Qwen3VLTextModel(cfg.text, bk)
        self.lm_head = nn.Linear(cfg.text.hidden_size, cfg.text.vocab_size,
                                 bias=False)
        if cfg.text.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.loss_fn = None

    # ---- float TMRoPE -----------------------------------------------------
    def get_rope_index(self, input_ids: torch.Tensor,
                       image_grid_thw: torch.Tensor | None = None,
                       video_grid_thw: torch.Tensor | None = None,
                       audio_seqlens: torch.Tensor | None = None,
                       second_per_grids: torch.Tensor | None = None,
                       use_audio_in_video: bool = False) -> torch.Tensor:
        cfg = self.config
        pps = cfg.position_id_per_seconds
        merge = cfg.vision.spatial_merge_size
        B, S = input_ids.shape
        pos = torch.zeros(3, B, S, dtype=torch.float, device=input_ids.device)
        img_i = vid_i = aud_i = 0
        for b in range(B):
            toks = input_ids[b].tolist()
            parts: list[torch.Tensor] = []

            def nxt() -> float:
                return float(parts[-1].max()) + 1 if parts else 0.0

            st = 0
            while st < len(toks):
                try:
                    ed_vs = toks.index(cfg.vision_start_token_id, st)
                except ValueError:
                    ed_vs = len(toks) + 1
                try:
                    ed_as = toks.index(cfg.audio_start_token_id, st)
                except ValueError:
                    ed_as = len(toks) + 1
                ed = min(ed_vs, ed_as)
                if ed > len(toks):
                    parts.append(_ftext(len(toks) - st, nxt()))
                    break
                st_idx = nxt()
                text_len = ed - st
                if text_len:
                    parts.append(_ftext(text_len, st_idx))
                    st_idx += text_len
                audio_in_video = (ed == ed_vs and ed_vs + 1 == ed_as)
                bos_len = 2 if audio_in_video else 1
                parts.append(_ftext(bos_len, st_idx))
                st_idx += bos_len
                if ed == ed_as and not audio_in_video:
                    alen = int(aut_output_lengths(audio_seqlens[aud_i],
                                                  cfg.audio.n_window))
                    parts.append(_ftext(alen, st_idx))
                    st += text_len + bos_len + alen + 1
                    aud_i += 1
                elif audio_in_video:
                    alen = int(aut_output_lengths(audio_seqlens[aud_i],
                                                  cfg.audio.n_window))
                    a_pos = _ftext(alen, st_idx)
                    t, h, w = video_grid_thw[vid_i].tolist()
                    spg = (float(second_per_grids[vid_i])
                           if second_per_grids is not None else 1.0)
                    t_idx = torch.arange(t).float() * spg * pps
                    v_pos = _fvision(st_idx, t_idx, h // merge, w // merge)
                    vi = ai = 0
                    while vi < v_pos.shape[1] and ai < a_pos.shape[1]:
                        if float(v_pos[0, vi]) <= float(a_pos[0, ai]):
                            parts.append(v_pos[:, vi:vi + 1])
                            vi += 1
                        else:
                            parts.append(a_pos[:, ai:ai + 1])
                            ai += 1
                    if vi < v_pos.shape[1]:
                        parts.append(v_pos[:, vi:])
                    if ai < a_pos.shape[1]:
                        parts.append(a_pos[:, ai:])
                    vlen = t * (h // merge) * (w // merge)
                    st += text_len + bos_len + alen + vlen + 2
                    aud_i += 1
                    vid_i += 1
                elif toks[ed_vs + 1] == cfg.image_token_id:
                    t, h, w = image_grid_thw[img_i].tolist()
                    t_idx = torch.arange(t).float() * pps
                    parts.append(_fvision(st_idx, t_idx, h // merge, w // merge))
                    st += text_len + bos_len + t * (h // merge) * (w // merge) + 1
                    img_i += 1
                else:
                    t, h, w = video_grid_thw[vid_i].tolist()
                    spg = (float(second_per_grids[vid_i])
                           if second_per_grids is not None else 1.0)
                    t_idx = torch.arange(t).float() * spg * pps
                    parts.append(_fvision(st_idx, t_idx, h // merge, w // merge))
                    st += text_len + bos_len + t * (h // merge) * (w // merge) + 1
                    vid_i += 1
                eos_len = 2 if audio_in_video else 1
                parts.append(_ftext(eos_len, nxt()))
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
                position_ids: torch.Tensor | None = None,
                return_hidden: bool = False, **_: Any):
        cfg = self.config
        x = self.model.embed_tokens(input_ids)
        audio_lens = None
        if input_features is not None:
            if feature_attention_mask is not None:
                audio_lens = feature_attention_mask.sum(-1)
                packed = input_features.permute(0, 2, 1)[
                    feature_attention_mask.bool()].T
            else:
                audio_lens = torch.tensor(
                    [input_features.shape[-1]] * input_features.shape[0],
                    device=input_ids.device)
                packed = input_features.permute(1, 0, 2).reshape(
                    input_features.shape[1], -1)
            audio = self.audio_tower(packed.to(x.dtype), audio_lens)
            amask = input_ids == cfg.audio_token_id
            x = x.clone()
            x[amask] = audio.to(x.dtype)
        vis_mask = None
        deepstack = None
        if pixel_values is not None:
            img, ds_img = self.visual(pixel_values.to(x.dtype), image_grid_thw)
            imask = input_ids == cfg.image_token_id
            x = x.clone()
            x[imask] = img.to(x.dtype)
            vis_mask, deepstack = imask, ds_img
        if pixel_values_videos is not None:
            vid, ds_vid = self.visual(pixel_values_videos.to(x.dtype),
                                      video_grid_thw)
            vmask = input_ids == cfg.video_token_id
            x = x.clone()
            x[vmask] = vid.to(x.dtype)
            if vis_mask is None:
                vis_mask, deepstack = vmask, ds_vid
            else:
                # join image + video deepstack features in token order
                joint_mask = vis_mask | vmask
                joined = []
                for di, dv in zip(deepstack, ds_vid):
                    buf = di.new_zeros(int(joint_mask.sum()), di.shape[-1])
                    buf[vis_mask[joint_mask]] = di
                    buf[vmask[joint_mask]] = dv
                    joined.append(buf)
                vis_mask, deepstack = joint_mask, joined
        if position_ids is None or position_ids.dim() != 3:
            position_ids = self.get_rope_index(
                input_ids, image_grid_thw, video_grid_thw, audio_lens,
                video_second_per_grid, use_audio_in_video)
        h = self.model(x, position_ids, visual_pos_mask=vis_mask,
                       deepstack=deepstack)
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

    def freeze_towers(self) -> None:
        for p in (*self.audio_tower.parameters(), *self.visual.parameters()):
            p.requires_grad_(False)

    @torch.no_grad()
    def init_weights(self, device=None) -> None:
        if device is not None:
            self.to_empty(device=device)
            a = self.config.audio
            self.audio_tower.positional_embedding.copy_(
                _sinusoids(a.max_source_positions, a.d_model).to(
                    self.audio_tower.positional_embedding.device))
            tc = self.config.text
            inv = 1.0 / (tc.rope_theta ** (torch.arange(
                0, tc.head_dim, 2, dtype=torch.float32) / tc.head_dim))
            self.model.inv_freq.copy_(inv.to(self.model.inv_freq.device))
            vc = self.config.vision
            hd = vc.hidden_size // vc.num_heads
            vinv = 1.0 / (10000.0 ** (torch.arange(0, hd // 2, 2).float()
                                      / (hd // 2)))
            self.visual.rot_inv_freq.copy_(vinv.to(self.visual.rot_inv_freq.device))
        std = self.config.initializer_range
        for m in self.modules():
            if isinstance(m, (nn.Linear, nn.Conv2d, nn.Conv3d)):
                nn.init.normal_(m.weight, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, (nn.LayerNorm,)):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
            elif isinstance(m, RMSNorm) or type(m).__name__ == "RMSNorm":
                nn.init.ones_(m.weight)
            elif type(m).__name__ == "MoE":
                nn.init.normal_(m.gate.weight, std=std)
                nn.init.normal_(m.experts.gate_proj, std=std)
                nn.init.normal_(m.experts.up_proj, std=std)
                nn.init.normal_(m.experts.down_proj, std=std)
        if self.config.text.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
