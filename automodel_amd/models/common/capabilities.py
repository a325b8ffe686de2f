"""Model capability flags + mesh validation.

Reference behavior: nemo_automodel/_transformers/capabilities.py:583
(per-model capability flags validated against the mesh before training).
"""

from __future__ import annotations

from dataclasses import dataclass


@dataclass
class ModelCapabilities:
    supports_tp: bool = True
    supports_pp: bool = True
    supports_cp: bool = True
    supports_ep: bool = False
    supports_packed_sequences: bool = True
    supports_lora: bool = True
    flash_head_dims: tuple[int, ...] = (128,)


_CAPS = {
    "LlamaForCausalLM": ModelCapabilities(),
    "MoEForCausalLM": ModelCapabilities(supports_ep=True, supports_cp=False,
                                       supports_tp=True),
    "DeepseekV3ForCausalLM": ModelCapabilities(supports_ep=True, supports_tp=False,
                                               supports_cp=False, flash_head_dims=()),
    "VLMForConditionalGeneration": ModelCapabilities(supports_pp=False,
                                                     supports_cp=False),
    "LlamaForSequenceClassification": ModelCapabilities(supports_pp=False,
                                                        supports_cp=False),
    # families without a TP plan / with non-flash attention paths
    # PP stages replay a llama-shaped forward; gemma needs embed scaling +
    # dual-frequency rope, so PP is not supported for the family
    "GemmaForCausalLM": ModelCapabilities(supports_cp=False,
                                          supports_pp=False, flash_head_dims=()),
    "Gemma3ForCausalLM": ModelCapabilities(supports_cp=False,
                                           supports_pp=False, flash_head_dims=()),
    "Gemma3ForConditionalGeneration": ModelCapabilities(
        supports_tp=False, supports_cp=False, supports_pp=False,
        flash_head_dims=()),
    "GptOssForCausalLM": ModelCapabilities(supports_tp=False, supports_cp=False,
                                           supports_ep=False, flash_head_dims=()),
    "NemotronForCausalLM": ModelCapabilities(supports_cp=False,
                                             flash_head_dims=()),
    "Glm4MoeForCausalLM": ModelCapabilities(supports_tp=False,
                                            supports_cp=False,
                                            flash_head_dims=()),
    "Qwen2VLForConditionalGeneration": ModelCapabilities(
        supports_tp=False, supports_cp=False, supports_pp=False),
    "Qwen2_5_VLForConditionalGeneration": ModelCapabilities(
        supports_tp=False, supports_cp=False, supports_pp=False),
    "LlavaForConditionalGeneration": ModelCapabilities(
        supports_tp=False, supports_cp=False, supports_pp=False),
    "Mistral3ForConditionalGeneration": ModelCapabilities(
        supports_tp=False, supports_cp=False, supports_pp=False),
    "NemotronHForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "BambaForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "GraniteMoeHybridForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "FalconH1ForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "Qwen3NextForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "Lfm2ForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "JambaForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "Zamba2ForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "MambaForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "Mamba2ForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "FalconMambaForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "RecurrentGemmaForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "PhimoeForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_ep=True, flash_head_dims=()),
    "GPTBigCodeForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        flash_head_dims=()),
    "Cohere2ForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        flash_head_dims=()),
    "Qwen3VLForConditionalGeneration": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "Qwen3VLMoeForConditionalGeneration": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, supports_ep=True, flash_head_dims=()),
    "Glm4vForConditionalGeneration": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "Qwen2AudioForConditionalGeneration": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, flash_head_dims=()),
    "Llama4ForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False),
    # generic transformers fallback: DP/FSDP only (reference non-custom path)
    "HFFallbackForCausalLM": ModelCapabilities(
        supports_tp=False, supports_pp=False, supports_cp=False,
        supports_packed_sequences=False, supports_lora=False,
        flash_head_dims=()),
}


def get_capabilities(model) -> ModelCapabilities:
    return _CAPS.get(type(model).__name__, ModelCapabilities())


def validate_model_against_mesh(model, mesh_dims: dict) -> list[str]:
    """Returns a list of violations (empty = valid). The recipe raises on any."""
    caps = get_capabilities(model)
    problems = []
    if mesh_dims.get("tp", 1) > 1 and not caps.supports_tp:
        problems.append(f"{type(model).__name__} has no TP plan")
    if mesh_dims.get("pp", 1) > 1 and not caps.supports_pp:
        problems.append(f"{type(model).__name__} does not support PP")
    if mesh_dims.get("cp", 1) > 1 and not caps.supports_cp:
        problems.append(f"{type(model).__name__} does not support CP")
    cfg = getattr(model, "config", None)
    if mesh_dims.get("tp", 1) > 1 and (
        getattr(cfg, "fused_qkv", False) or getattr(cfg, "fused_gate_up", False)
    ):
        problems.append("fused qkv/gate_up projections have no TP plan (disable fusion)")
    if mesh_dims.get("cp", 1) > 1 and getattr(cfg, "sliding_window", None):
        problems.append("sliding-window attention has no CP mechanism (disable cp or the window)")
    heads = getattr(getattr(model, "config", None), "num_attention_heads", None)
    tp = mesh_dims.get("tp", 1)
    if heads and tp > 1 and heads % tp != 0:
        problems.append(f"num_attention_heads {heads} not divisible by tp={tp}")
    return problems
