"""DeepSeek-V3-style MLA model tests (CPU)."""

import torch

from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy
from automodel_amd.models.deepseek_v3.model import (
    DeepseekV3Config, DeepseekV3ForCausalLM,
)

TINY = dict(
    vocab_size=256, hidden_size=64, intermediate_size=128, num_hidden_layers=3,
    num_attention_heads=4, first_k_dense_replace=1, q_lora_rank=32,
    kv_lora_rank=16, qk_nope_head_dim=16, qk_rope_head_dim=8, v_head_dim=16,
    max_position_embeddings=64,
    moe={"n_routed_experts": 4, "n_shared_experts": 1, "n_activated_experts": 2,
         "score_func": "sigmoid", "expert_bias": True, "moe_intermediate_size": 32,
         "shared_expert_intermediate_size": 32},
)


def make_model():
    m = DeepseekV3ForCausalLM(TINY)
    m.init_weights()
    m.loss_fn = FusedLinearCrossEntropy(backend="chunked", chunk_size=32)
    return m


def test_mla_shapes_and_layer_mix():
    m = make_model()
    from automodel_amd.moe.layers import MoE
    from automodel_amd.models.deepseek_v3.model import DenseMLP

    assert isinstance(m.model.layers[0].mlp, DenseMLP)
    assert isinstance(m.model.layers[1].mlp, MoE)
    ids = torch.randint(0, 256, (2, 16))
    logits = m(ids)
    assert logits.shape == (2, 16, 256)


def test_mla_train_step_and_bias_update():
    torch.manual_seed(0)
    m = make_model()
    ids = torch.randint(0, 256, (2, 17))
    loss = m(ids[:, :-1].contiguous(), labels=ids[:, 1:].contiguous())
    assert torch.isfinite(loss)
    loss.backward()
    assert m.model.layers[0].self_attn.kv_b_proj.weight.grad is not None
    assert m.model.layers[1].mlp.experts.gate_proj.grad is not None
    bias_before = m.model.layers[1].mlp.gate.e_score_correction_bias.clone()
    m.update_moe_gate_bias()
    assert not torch.equal(bias_before, m.model.layers[1].mlp.gate.e_score_correction_bias)


def test_deepseek_hf_config_mapping():
    cfg = DeepseekV3Config.from_hf_config({
        "architectures": ["DeepseekV3ForCausalLM"], "vocab_size": 1000,
        "hidden_size": 128, "num_hidden_layers": 4, "num_attention_heads": 8,
        "q_lora_rank": 64, "kv_lora_rank": 32, "qk_nope_head_dim": 16,
        "qk_rope_head_dim": 8, "v_head_dim": 16, "n_routed_experts": 16,
        "num_experts_per_tok": 4, "scoring_func": "sigmoid",
        "routed_scaling_factor": 2.5, "first_k_dense_replace": 2,
    })
    assert cfg.moe.n_routed_experts == 16
    assert cfg.moe.route_scale == 2.5
    assert cfg.qk_head_dim == 24


def test_registry_builds_deepseek():
    from automodel_amd.models.registry import build_model

    m = build_model(config=TINY, architecture="DeepseekV3ForCausalLM",
                    dtype="float32", meta_init=False)
    assert type(m).__name__ == "DeepseekV3ForCausalLM"
