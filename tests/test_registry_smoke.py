"""Registry-wide smoke: every registered LM family instantiates from a tiny
config, init_weights()s, and runs a forward pass.

Reference behavior: nemo_automodel/_transformers/registry.py (architecture
registry); this guards against registration/`config_class` drift as
families are added. VLM ConditionalGeneration classes are covered by
their own tests (vision towers have large config defaults)."""

import dataclasses

import pytest
import torch

from automodel_amd.models.registry import _REGISTRY, _ensure_builtin

_SHRINK = dict(vocab_size=64, hidden_size=32, num_hidden_layers=2,
               intermediate_size=48, num_attention_heads=4,
               num_key_value_heads=2, head_dim=8,
               max_position_embeddings=64, max_seq_len=64,
               d_model=32, n_layers=2, n_heads=4, n_layer=2, n_head=4,
               n_embd=32, ffn_dim=48, n_inner=48, kv_n_heads=2,
               num_kv_heads=2, ffn_hidden_size=24, moe_num_experts=4,
               moe_top_k=2, rotary_dim=4, num_local_experts=4,
               num_experts_per_tok=2, num_experts=4, moe_topk=2,
               num_shared_expert=1, block_size=8)


def _lm_classes():
    _ensure_builtin()
    seen, out = set(), []
    for name, cls in sorted(_REGISTRY.items()):
        if cls in seen or "ConditionalGeneration" in name:
            continue
        seen.add(cls)
        out.append(pytest.param(cls, id=name))
    return out


@pytest.mark.parametrize("cls", _lm_classes())
def test_registered_family_builds_and_steps(cls):
    cfg_cls = cls.config_class
    assert dataclasses.is_dataclass(cfg_cls), cls
    fields = {f.name for f in dataclasses.fields(cfg_cls)}
    kw = {k: v for k, v in _SHRINK.items() if k in fields}
    torch.manual_seed(0)
    model = cls(cfg_cls(**kw))
    model.init_weights()
    ids = torch.randint(0, 64, (1, 12))
    with torch.no_grad():
        logits = model(ids)
    assert logits.shape[:2] == (1, 12)
    assert torch.isfinite(logits).all()
