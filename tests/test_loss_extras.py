"""Distill-extra losses, embedding row repair, keep-top-k retention,
capabilities CLI."""

import os

import pytest
import torch
import torch.nn as nn

from automodel_amd.loss.distill_extras import (
    EmbeddingDistillLoss,
    IntermediateDistillLoss,
    listmle_loss,
)


def test_embedding_distill_identical_is_zero():
    h = torch.randn(2, 8, 16)
    loss = EmbeddingDistillLoss(16, 16, mode="cosine")(h, h.clone())
    assert loss.abs() < 1e-6
    loss_mse = EmbeddingDistillLoss(16, 16, mode="mse")(h, h.clone())
    assert loss_mse.abs() < 1e-8


def test_embedding_distill_projection_and_mask():
    torch.manual_seed(0)
    l = EmbeddingDistillLoss(8, 16, mode="mse")
    s = torch.randn(2, 4, 8, requires_grad=True)
    t = torch.randn(2, 4, 16)
    mask = torch.tensor([[1., 1., 0., 0.], [1., 0., 0., 0.]])
    out = l(s, t, mask)
    out.backward()
    assert s.grad is not None
    assert s.grad[0, 2].abs().sum() == 0  # masked tokens contribute nothing


def test_intermediate_distill_layer_map():
    torch.manual_seed(1)
    l = IntermediateDistillLoss(8, 8, layer_map={0: 1, 1: 3}, mode="mse")
    s_layers = [torch.randn(1, 4, 8) for _ in range(2)]
    t_layers = [torch.randn(1, 4, 8) for _ in range(4)]
    v = l(s_layers, t_layers)
    ref = (((s_layers[0] - t_layers[1]) ** 2).mean()
           + ((s_layers[1] - t_layers[3]) ** 2).mean()) / 2
    torch.testing.assert_close(v, ref)


def test_listmle_prefers_correct_ranking():
    rel = torch.tensor([[3.0, 2.0, 1.0, 0.0]])
    good = torch.tensor([[5.0, 3.0, 1.0, -1.0]], requires_grad=True)
    bad = torch.tensor([[-1.0, 1.0, 3.0, 5.0]])
    assert listmle_loss(good, rel) < listmle_loss(bad, rel)
    listmle_loss(good, rel).backward()
    assert good.grad is not None


def test_repair_embedding_rows():
    from automodel_amd.training.extras import repair_embedding_rows

    e = nn.Embedding(10, 4)
    with torch.no_grad():
        e.weight[3] = float("nan")
        e.weight[7, 2] = float("inf")
    n = repair_embedding_rows(e)
    assert n == 2
    assert torch.isfinite(e.weight).all()
    assert repair_embedding_rows(e) == 0


def test_keep_top_k_retention(tmp_path):
    from automodel_amd.checkpoint.checkpointing import Checkpointer

    ck = Checkpointer(checkpoint_dir=str(tmp_path), keep_top_k=2)
    for step, metric in [(1, 0.9), (2, 0.5), (3, 0.7), (4, 0.8)]:
        os.makedirs(tmp_path / f"step_{step}")
        ck.record_metric(step, metric)   # lower is better by default
    ck._apply_retention()
    left = sorted(os.listdir(tmp_path))
    # best two metrics: step_2 (0.5), step_3 (0.7); latest step_4 protected
    assert left == ["step_2", "step_3", "step_4"]


def test_capabilities_cli(capsys):
    from automodel_amd.cli.app import query_capabilities

    query_capabilities("DeepseekV3ForCausalLM")
    out = capsys.readouterr().out
    assert "supports_ep=True" in out and "supports_tp=False" in out
    query_capabilities()  # all architectures
    assert "GptOssForCausalLM" in capsys.readouterr().out


def test_evaluate_nll():
    from automodel_amd.eval.perplexity import evaluate_nll
    from automodel_amd.models.llama.model import LlamaForCausalLM

    torch.manual_seed(0)
    m = LlamaForCausalLM(dict(vocab_size=100, hidden_size=32, intermediate_size=64,
                              num_hidden_layers=2, num_attention_heads=2,
                              num_key_value_heads=1, max_position_embeddings=64))
    m.init_weights(device="cpu")
    batches = [{"input_ids": torch.randint(0, 100, (2, 17))} for _ in range(3)]
    out = evaluate_nll(m, batches)
    assert out["n_tokens"] == 3 * 2 * 16
    assert out["perplexity"] > 1.0
    # random-init model on uniform-random tokens: nll near ln(vocab)
    import math

    assert abs(out["nll_per_token"] - math.log(100)) < 1.0
    # ignore_index masking respected
    b = {"input_ids": torch.randint(0, 100, (1, 9)),
         "labels": torch.full((1, 9), -100)}
    out2 = evaluate_nll(m, [b])
    assert out2["n_tokens"] == 0
