"""Tensor/sequence parallelism: DTensor parallelize_module plans.

Reference behavior: nemo_automodel/components/distributed/optimized_tp_plans.py
:174-765 (per-model colwise/rowwise plans + sequence-parallel styles) and
parallelizer.py:2126 (_get_parallel_plan: explicit YAML plan > model-optimized
plan > HF tp_plan translation).

MI355X note: TP collectives ride RCCL over xGMI point-to-point links; row-
parallel all-reduce of [B,S,H] activations is the dominant message — keep TP
degree low (2-4) on a single node and prefer SP (all-gather + reduce-scatter)
for long sequences.
"""

from __future__ import annotations

import torch
import torch.nn as nn
from torch.distributed.device_mesh import DeviceMesh
from torch.distributed.tensor import Replicate, Shard
from torch.distributed.tensor.parallel import (
    ColwiseParallel,
    RowwiseParallel,
    SequenceParallel,
    parallelize_module,
)


def llama_tp_plan(sequence_parallel: bool = False) -> dict:
    """Per-layer plan for the Llama family (reference optimized_tp_plans.py:174)."""
    plan: dict = {
        "model.layers.*.self_attn.q_proj": ColwiseParallel(),
        "model.layers.*.self_attn.k_proj": ColwiseParallel(),
        "model.layers.*.self_attn.v_proj": ColwiseParallel(),
        "model.layers.*.self_attn.o_proj": RowwiseParallel(),
        "model.layers.*.mlp.gate_proj": ColwiseParallel(),
        "model.layers.*.mlp.up_proj": ColwiseParallel(),
        "model.layers.*.mlp.down_proj": RowwiseParallel(),
    }
    if sequence_parallel:
        plan.update({
            "model.layers.*.input_layernorm": SequenceParallel(),
            "model.layers.*.post_attention_layernorm": SequenceParallel(),
            "model.norm": SequenceParallel(),
            "model.layers.*.self_attn.q_proj": ColwiseParallel(input_layouts=Shard(1)),
            "model.layers.*.self_attn.k_proj": ColwiseParallel(input_layouts=Shard(1)),
            "model.layers.*.self_attn.v_proj": ColwiseParallel(input_layouts=Shard(1)),
            "model.layers.*.self_attn.o_proj": RowwiseParallel(output_layouts=Shard(1)),
            "model.layers.*.mlp.gate_proj": ColwiseParallel(input_layouts=Shard(1)),
            "model.layers.*.mlp.up_proj": ColwiseParallel(input_layouts=Shard(1)),
            "model.layers.*.mlp.down_proj": RowwiseParallel(output_layouts=Shard(1)),
            "model.embed_tokens": RowwiseParallel(
                input_layouts=Replicate(), output_layouts=Shard(1)
            ),
            "lm_head": ColwiseParallel(input_layouts=Shard(1), output_layouts=Replicate()),
        })
    return plan


def gemma_tp_plan(sequence_parallel: bool = False) -> dict:
    """Gemma family: same projection layout as llama; SP would need plans
    for the four per-layer norms — not wired yet."""
    assert not sequence_parallel, "sequence_parallel not supported for gemma"
    return llama_tp_plan(False)


def nemotron_tp_plan(sequence_parallel: bool = False) -> dict:
    assert not sequence_parallel, "sequence_parallel not supported for nemotron"
    return {
        "model.layers.*.self_attn.q_proj": ColwiseParallel(),
        "model.layers.*.self_attn.k_proj": ColwiseParallel(),
        "model.layers.*.self_attn.v_proj": ColwiseParallel(),
        "model.layers.*.self_attn.o_proj": RowwiseParallel(),
        "model.layers.*.mlp.up_proj": ColwiseParallel(),
        "model.layers.*.mlp.down_proj": RowwiseParallel(),
    }


def moe_attention_tp_plan(sequence_parallel: bool = False) -> dict:
    """MoE families: shard attention only; the stacked expert parameters and
    the router stay replicated (EP is the expert-scaling axis — reference
    parallelizer.py:219 safe replicated-router plan)."""
    assert not sequence_parallel, "sequence_parallel not supported for MoE TP"
    return {
        "model.layers.*.self_attn.q_proj": ColwiseParallel(),
        "model.layers.*.self_attn.k_proj": ColwiseParallel(),
        "model.layers.*.self_attn.v_proj": ColwiseParallel(),
        "model.layers.*.self_attn.o_proj": RowwiseParallel(),
    }


_MODEL_PLANS = {
    "LlamaForCausalLM": llama_tp_plan,
    "Qwen2ForCausalLM": llama_tp_plan,
    "MistralForCausalLM": llama_tp_plan,
    "GemmaForCausalLM": gemma_tp_plan,
    "Gemma3ForCausalLM": gemma_tp_plan,
    "NemotronForCausalLM": nemotron_tp_plan,
    "MoEForCausalLM": moe_attention_tp_plan,
}


def get_tp_plan(model: nn.Module, sequence_parallel: bool = False,
                explicit_plan: dict | None = None) -> dict:
    """explicit plan > model-optimized plan (reference parallelizer.py:2126)."""
    if explicit_plan:
        return explicit_plan
    name = type(model).__name__
    for arch, fn in _MODEL_PLANS.items():
        if name == arch or arch in getattr(model, "hf_architectures", ()):
            return fn(sequence_parallel)
    raise KeyError(f"no TP plan registered for {name}")


def _rewrite_head_counts(model: nn.Module, tp_size: int) -> None:
    """After head-sharding, per-rank module head counts shrink (reference
    parallelizer.py:1558)."""
    for m in model.modules():
        if hasattr(m, "num_heads") and hasattr(m, "num_kv_heads"):
            assert m.num_heads % tp_size == 0 and m.num_kv_heads % tp_size == 0, (
                f"heads ({m.num_heads}/{m.num_kv_heads}) not divisible by tp={tp_size}"
            )
            m.num_heads //= tp_size
            m.num_kv_heads //= tp_size


def apply_tp(
    model: nn.Module,
    tp_mesh: DeviceMesh,
    sequence_parallel: bool = False,
    plan: dict | None = None,
) -> nn.Module:
    tp_size = tp_mesh.size()
    if tp_size == 1:
        return model
    resolved = get_tp_plan(model, sequence_parallel, plan)
    parallelize_module(model, tp_mesh, resolved)
    _rewrite_head_counts(model, tp_size)
    return model
