"""Sequence-classification recipe (reference: recipes/llm/train_seq_cls.py:474).

Backbone + pooled classification head; labels are per-sequence class ids.
"""

from __future__ import annotations

import sys

import torch
import torch.nn as nn

from automodel_amd.config.loader import apply_overrides, load_yaml_config, parse_cli_overrides
from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM
from automodel_amd.recipes.llm.train_ft import TrainFinetuneRecipeForNextTokenPrediction


class LlamaForSequenceClassification(nn.Module):
    config_class = LlamaConfig

    def __init__(self, config, num_labels: int = 2, backend=None):
        super().__init__()
        if isinstance(config, dict):
            config = LlamaConfig(**config)
        self.num_labels = num_labels
        base = LlamaForCausalLM(config, backend=backend)
        self.model = base.model
        self.config = config
        self.score = nn.Linear(config.hidden_size, num_labels, bias=False)
        self.loss_fn = None  # unused; recipe computes CE over classes

    def forward(self, input_ids, labels=None, **_):
        hidden = self.model(input_ids)          # [B,S,H]
        pooled = hidden[:, -1]                  # last-token pooling
        logits = self.score(pooled)
        if labels is not None:
            return torch.nn.functional.cross_entropy(
                logits.float(), labels.reshape(-1), reduction="sum")
        return logits

    @torch.no_grad()
    def init_weights(self, device=None):
        from automodel_amd.ops.rms_norm import RMSNorm
        from automodel_amd.ops.rope import build_rope_cache

        if device is not None:
            self.to_empty(device=device)
            cos, sin = build_rope_cache(self.config.head_dim,
                                        self.config.max_position_embeddings,
                                        self.config.rope_theta,
                                        self.config.rope_scaling, device=device)
            self.model.rope_cos.copy_(cos)
            self.model.rope_sin.copy_(sin)
        std = self.config.initializer_range
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=std)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, RMSNorm):
                nn.init.ones_(m.weight)


class TrainFinetuneRecipeForSequenceClassification(TrainFinetuneRecipeForNextTokenPrediction):
    def setup(self) -> None:
        num_labels = self.cfg.get_by_dotted("model.num_labels", 2)
        self.cfg.model["_target_"] = (
            "automodel_amd.recipes.llm.train_seq_cls.build_seq_cls_model"
        )
        super().setup()

    def _forward_backward_step(self, batch, loss_scale):
        input_ids = batch["input_ids"].to(self.device, non_blocking=True)
        labels = batch["labels"].to(self.device, non_blocking=True)
        if labels.dim() > 1:            # class id per sequence
            labels = labels[:, 0]
        loss = self.model(input_ids, labels=labels)
        (loss * loss_scale).backward()
        return loss.detach()


def build_seq_cls_model(config=None, num_labels: int = 2, dtype: str = "float32", **_):
    import torch as _t

    cfg = dict(config) if config is not None else {}
    m = LlamaForSequenceClassification(cfg, num_labels=num_labels)
    return m.to(getattr(_t, dtype))


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    cfg = load_yaml_config(argv[0])
    apply_overrides(cfg, parse_cli_overrides(argv[1:]))
    r = TrainFinetuneRecipeForSequenceClassification(cfg)
    r.setup()
    r.run_train_validation_loop()


if __name__ == "__main__":
    main()
