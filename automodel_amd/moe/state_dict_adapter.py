"""HF per-expert keys <-> stacked expert tensors.

Reference behavior: nemo_automodel/components/moe/state_dict_mixin.py (953
LoC: HF per-expert keys <-> stacked DTensor expert weights, EP-aware).

Internal layout: model.layers.N.mlp.experts.{gate_proj,up_proj,down_proj}
as stacked [E, out, in] parameters and model.layers.N.mlp.gate.weight.

HF layouts:
  qwen3_moe: model.layers.N.mlp.experts.E.{gate_proj,up_proj,down_proj}.weight
             model.layers.N.mlp.gate.weight
  mixtral:   model.layers.N.block_sparse_moe.experts.E.{w1,w3,w2}.weight
             model.layers.N.block_sparse_moe.gate.weight
             (w1=gate, w3=up, w2=down)
"""

from __future__ import annotations

import re

import torch

_MIXTRAL_MAP = {"w1": "gate_proj", "w3": "up_proj", "w2": "down_proj"}
_MIXTRAL_INV = {v: k for k, v in _MIXTRAL_MAP.items()}


class MoEStateDictAdapter:
    def __init__(self, config):
        self.config = config
        self.flavor = getattr(config, "hf_flavor", "qwen3_moe")
        self.n_experts = config.moe.n_routed_experts

    def hf_key_targets(self, key: str) -> list[str]:
        """Internal param name(s) an HF key feeds — lets the streaming HF
        loader (checkpoint/hf_loader.py) free consumed shard keys."""
        m = re.match(r"^(model\.layers\.\d+)\.(?:mlp|block_sparse_moe)"
                     r"\.experts\.(\d+)\.(\w+)\.weight$", key)
        if m:
            proj = _MIXTRAL_MAP.get(m.group(3), m.group(3))
            return [f"{m.group(1)}.mlp.experts.{proj}"]
        if key.endswith("experts.gate_up_proj"):
            base = key.replace(".block_sparse_moe.", ".mlp.")
            return [base.replace("gate_up_proj", "gate_proj"),
                    base.replace("gate_up_proj", "up_proj")]
        if key.endswith(".gate.moe_statics.e_score_correction_bias"):
            return [key.replace(".gate.moe_statics.e_score_correction_bias",
                                ".gate.e_score_correction_bias")]
        if key.endswith(".block_sparse_moe.e_score_correction_bias") or \
                key.endswith(".mlp.e_score_correction_bias"):
            # MiniMax-M2 keeps the aux-free bias at the MoE-block level
            return [key.rsplit(".", 1)[0] + ".gate.e_score_correction_bias"]
        if ".mlp.shared_expert." in key:
            return [key.replace(".mlp.shared_expert.", ".mlp.shared_experts.")]
        if ".block_sparse_moe.router." in key:
            return [key.replace(".block_sparse_moe.router.", ".mlp.gate.")]
        if ".block_sparse_moe.gate." in key:
            return [key.replace(".block_sparse_moe.gate.", ".mlp.gate.")]
        if ".block_sparse_moe." in key:
            return [key.replace(".block_sparse_moe.", ".mlp.")]
        return [key]

    # ---- HF -> internal (stack per-expert tensors)
    def from_hf(self, sd: dict[str, torch.Tensor]) -> dict[str, torch.Tensor]:
        out: dict[str, torch.Tensor] = {}
        pending: dict[str, dict[int, torch.Tensor]] = {}
        pat_q = re.compile(r"^(model\.layers\.\d+)\.mlp\.experts\.(\d+)\.(gate_proj|up_proj|down_proj)\.weight$")
        pat_m = re.compile(r"^(model\.layers\.\d+)\.block_sparse_moe\.experts\.(\d+)\.(w1|w2|w3)\.weight$")
        for key, t in sd.items():
            m = pat_q.match(key) or pat_m.match(key)
            if m:
                layer, e, proj = m.group(1), int(m.group(2)), m.group(3)
                proj = _MIXTRAL_MAP.get(proj, proj)
                pending.setdefault(f"{layer}.mlp.experts.{proj}", {})[e] = t
            elif key.endswith("experts.gate_up_proj"):
                # transformers >= 4.56 stacked-expert layout: [E, 2I, H]
                # fused as [gate; up] along dim 1 (chunk(2, dim=1));
                # granite-moe stores them under block_sparse_moe
                base_key = key.replace(".block_sparse_moe.", ".mlp.")
                gate, up = t.chunk(2, dim=1)
                out[base_key.replace("gate_up_proj", "gate_proj")] = gate.contiguous()
                out[base_key.replace("gate_up_proj", "up_proj")] = up.contiguous()
            elif key.endswith(".gate.moe_statics.e_score_correction_bias"):
                # ernie: [1, E] fp32 selection bias -> flat buffer
                out[key.replace(".gate.moe_statics.e_score_correction_bias",
                                ".gate.e_score_correction_bias")] = t.reshape(-1)
            elif (key.endswith(".mlp.e_score_correction_bias")
                  or key.endswith(".block_sparse_moe.e_score_correction_bias")):
                # minimax-m2: block-level aux-free bias -> gate buffer
                out[key.rsplit(".", 1)[0].replace(".block_sparse_moe", ".mlp")
                    + ".gate.e_score_correction_bias"] = t.reshape(-1)
            elif ".mlp.shared_expert." in key:
                out[key.replace(".mlp.shared_expert.", ".mlp.shared_experts.")] = t
            elif ".block_sparse_moe.router." in key:   # granite-moe
                out[key.replace(".block_sparse_moe.router.", ".mlp.gate.")] = t
            elif ".block_sparse_moe.gate." in key:
                out[key.replace(".block_sparse_moe.gate.", ".mlp.gate.")] = t
            elif ".block_sparse_moe." in key:
                out[key.replace(".block_sparse_moe.", ".mlp.")] = t
            else:
                out[key] = t
        for stacked_key, parts in pending.items():
            out[stacked_key] = torch.stack([parts[e] for e in sorted(parts)], dim=0)
        return out

    # ---- internal -> HF (unstack)
    def to_hf(self, sd: dict[str, torch.Tensor]) -> dict[str, torch.Tensor]:
        out: dict[str, torch.Tensor] = {}
        if self.flavor == "qwen2_moe":
            sd = {k.replace(".mlp.shared_experts.", ".mlp.shared_expert."): v
                  for k, v in sd.items()}
        pat = re.compile(r"^(model\.layers\.\d+)\.mlp\.experts\.(gate_proj|up_proj|down_proj)$")
        if self.flavor == "ernie":
            # HF ernie stores experts STACKED: recombine gate+up -> gate_up
            for key, t in sd.items():
                m = pat.match(key)
                if m and m.group(2) == "gate_proj":
                    up = sd[key.replace("gate_proj", "up_proj")]
                    out[key.replace("gate_proj", "gate_up_proj")] = \
                        torch.cat([t, up], dim=1)
                elif m and m.group(2) == "up_proj":
                    continue
                elif key.endswith(".gate.e_score_correction_bias"):
                    out[key.replace(".gate.e_score_correction_bias",
                                    ".gate.moe_statics.e_score_correction_bias")] = \
                        t.reshape(1, -1)
                else:
                    out[key] = t
            return out
        for key, t in sd.items():
            m = pat.match(key)
            if m:
                layer, proj = m.group(1), m.group(2)
                for e in range(t.shape[0]):
                    if self.flavor == "mixtral":
                        out[f"{layer}.block_sparse_moe.experts.{e}.{_MIXTRAL_INV[proj]}.weight"] = t[e]
                    else:
                        out[f"{layer}.mlp.experts.{e}.{proj}.weight"] = t[e]
            elif self.flavor == "mixtral" and ".mlp.gate." in key:
                out[key.replace(".mlp.gate.", ".block_sparse_moe.gate.")] = t
            else:
                out[key] = t
        return out
