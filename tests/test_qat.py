"""QAT fake-quant: STE gradients, per-group/per-token quant error, prepare
swap, delayed enable, convert-to-int4 export, recipe integration."""

import pytest
import torch
import torch.nn as nn

from automodel_amd.quantization.qat import (
    QATConfig,
    QATLinear,
    fake_quant_per_group,
    fake_quant_per_token,
    maybe_enable_delayed_fake_quant,
    prepare_qat,
    set_fake_quant,
)


def test_fake_quant_per_group_values_on_grid():
    torch.manual_seed(0)
    w = torch.randn(8, 64)
    q = fake_quant_per_group(w, n_bits=4, group_size=32)
    # every value must be k * (group absmax / 7) for integer k in [-8, 7]
    g = q.reshape(8, 2, 32)
    scale = w.reshape(8, 2, 32).abs().amax(-1, keepdim=True) / 7
    k = g / scale
    torch.testing.assert_close(k, k.round(), atol=1e-4, rtol=1e-4)
    assert k.max() <= 7.01 and k.min() >= -8.01
    # error bounded by half a step
    assert (q - w).abs().max() <= (scale.max() / 2) + 1e-6


def test_fake_quant_ste_gradient_is_identity():
    w = torch.randn(4, 32, requires_grad=True)
    q = fake_quant_per_group(w, 4, 32)
    q.sum().backward()
    torch.testing.assert_close(w.grad, torch.ones_like(w))

    x = torch.randn(3, 16, requires_grad=True)
    fake_quant_per_token(x).sum().backward()
    torch.testing.assert_close(x.grad, torch.ones_like(x))


def test_qat_linear_forward_close_to_base():
    torch.manual_seed(1)
    base = nn.Linear(64, 32)
    q = QATLinear(base, QATConfig())
    x = torch.randn(5, 64)
    y_base = base(x)
    y_q = q(x)
    # int8 act + int4 weight fake quant: close but not equal
    assert not torch.equal(y_base, y_q)
    rel = (y_base - y_q).norm() / y_base.norm()
    assert rel < 0.15, rel.item()


def test_qat_weight_only_mode_ignores_activation():
    base = nn.Linear(32, 8)
    q = QATLinear(base, QATConfig(quantizer_type="int4_weight_only"))
    x = torch.randn(2, 32)
    w = fake_quant_per_group(q.weight, 4, 32)
    torch.testing.assert_close(q(x), torch.nn.functional.linear(x, w, q.bias))


def test_prepare_qat_skips_lm_head_and_trains():
    model = nn.ModuleDict({
        "up": nn.Linear(32, 32),
        "lm_head": nn.Linear(32, 8),
    })
    n = prepare_qat(model, {"group_size": 16})
    assert n == 1 and isinstance(model["up"], QATLinear)
    assert type(model["lm_head"]) is nn.Linear
    x = torch.randn(2, 32)
    model["lm_head"](model["up"](x)).sum().backward()
    assert model["up"].weight.grad is not None


def test_delayed_fake_quant():
    base = nn.Linear(32, 8)
    m = nn.Sequential(QATLinear(base, QATConfig(delay_steps=5)))
    assert not m[0].fake_quant_enabled
    x = torch.randn(2, 32)
    torch.testing.assert_close(m(x), base(x))  # full precision before delay
    maybe_enable_delayed_fake_quant(m, step=3)
    assert not m[0].fake_quant_enabled
    maybe_enable_delayed_fake_quant(m, step=5)
    assert m[0].fake_quant_enabled
    set_fake_quant(m, False)
    assert not m[0].fake_quant_enabled


def test_convert_export_roundtrip():
    torch.manual_seed(2)
    q = QATLinear(nn.Linear(64, 16), QATConfig())
    codes, scales = q.convert()
    assert codes.dtype == torch.int8 and codes.shape == (16, 64)
    assert scales.shape == (16, 2)
    recon = (codes.float().reshape(16, 2, 32) * scales[..., None]).reshape(16, 64)
    ref = fake_quant_per_group(q.weight.detach(), 4, 32)
    torch.testing.assert_close(recon, ref.float(), atol=1e-5, rtol=1e-4)


def test_qat_config_validation():
    with pytest.raises(ValueError):
        QATConfig(quantizer_type="fp1")
    c = QATConfig.from_config({"quantizer_type": "int4_weight_only",
                               "group_size": 16, "enabled": True})
    assert c.group_size == 16


def test_qat_recipe_integration(tmp_path):
    from automodel_amd.config.loader import ConfigNode
    from automodel_amd.recipes.llm.train_ft import (
        TrainFinetuneRecipeForNextTokenPrediction,
    )

    cfg = ConfigNode({
        "seed": 0,
        "model": {"config": {
            "vocab_size": 128, "hidden_size": 32, "intermediate_size": 64,
            "num_hidden_layers": 2, "num_attention_heads": 2,
            "num_key_value_heads": 1, "max_position_embeddings": 64,
        }, "dtype": "float32"},
        "qat": {"group_size": 16},
        "loss_fn": {"backend": "chunked", "chunk_size": 64},
        "optimizer": {"lr": 1e-3},
        "step_scheduler": {"max_steps": 2},
        "dataloader": {"dataset": {"kind": "mock", "num_samples": 8,
                                   "seq_len": 32, "vocab_size": 128},
                       "batch_size": 2},
        "output_dir": str(tmp_path),
    })
    r = TrainFinetuneRecipeForNextTokenPrediction(cfg)
    r.setup()
    n_qat = sum(1 for m in r.model.modules() if isinstance(m, QATLinear))
    assert n_qat > 0
    r.run_train_validation_loop()


def test_prepare_qat_skips_lora_adapters():
    from automodel_amd.peft.lora import LinearLoRA

    m = nn.ModuleDict({"q_proj": LinearLoRA(nn.Linear(32, 32), dim=4, alpha=8)})
    n = prepare_qat(m, {"group_size": 16})
    # base linear inside the LoRA wrapper is quantized; adapters are NOT
    assert n == 1
    assert isinstance(m["q_proj"].base, QATLinear)
    assert type(m["q_proj"].lora_A) is nn.Linear
