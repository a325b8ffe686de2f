"""Golden-loss regression (reference CI pattern: golden value JSONL per
config — SURVEY §4 'CI recipe validation'). Regenerate deliberately with
tests/golden_values README procedure when numerics change on purpose."""

import json
import os

import torch


def test_tiny_llama_golden_loss_trajectory():
    from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy
    from automodel_amd.models.common.backend import BackendConfig
    from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM
    from automodel_amd.optim.adamw import FusedAdamW

    golden = json.load(open(os.path.join(os.path.dirname(__file__),
                                         "golden_values/tiny_llama_cpu.json")))
    torch.manual_seed(golden["seed"])
    cfg = LlamaConfig(vocab_size=256, hidden_size=64, intermediate_size=128,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=2, max_position_embeddings=64)
    m = LlamaForCausalLM(cfg, backend=BackendConfig().for_cpu())
    m.init_weights()
    m.loss_fn = FusedLinearCrossEntropy(backend="chunked", chunk_size=64)
    opt = FusedAdamW(m.parameters(), lr=1e-3, weight_decay=0.0)
    g = torch.Generator().manual_seed(golden["data_seed"])
    for step, expected in enumerate(golden["losses"]):
        ids = torch.randint(0, 256, (2, 33), generator=g)
        loss = m(ids[:, :-1].contiguous(), labels=ids[:, 1:].contiguous()) / 64
        opt.zero_grad(); loss.backward(); opt.step()
        assert abs(float(loss.detach()) - expected) < 5e-4, \
            (step, float(loss.detach()), expected)
