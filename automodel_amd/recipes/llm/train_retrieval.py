"""Retrieval bi-encoder training recipe (query/passage InfoNCE).

Reference behavior: nemo_automodel/recipes/retrieval/ (bi-encoder training
with cross-rank in-batch negatives). The encoder is a Llama backbone with
mean pooling; loss is InfoNCE over the DP-global batch.
"""

from __future__ import annotations

import sys

import torch
import torch.nn as nn

from automodel_amd.config.loader import ConfigNode, apply_overrides, load_yaml_config, parse_cli_overrides
from automodel_amd.loss.infonce import info_nce_loss
from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM
from automodel_amd.recipes.llm.train_ft import TrainFinetuneRecipeForNextTokenPrediction


class BiEncoder(nn.Module):
    """Shared-weight encoder with mean pooling (reference retrieval.py)."""

    config_class = LlamaConfig

    def __init__(self, config, backend=None, embedding_dim: int | None = None):
        super().__init__()
        if isinstance(config, dict):
            config = LlamaConfig(**config)
        self.config = config
        base = LlamaForCausalLM(config, backend=backend)
        self.model = base.model
        self.proj = (nn.Linear(config.hidden_size, embedding_dim, bias=False)
                     if embedding_dim else nn.Identity())
        self.loss_fn = None  # recipe computes InfoNCE

    def encode(self, input_ids: torch.Tensor) -> torch.Tensor:
        hidden = self.model(input_ids)
        return self.proj(hidden.mean(dim=1))

    def forward(self, input_ids, **_):
        return self.encode(input_ids)

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
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, RMSNorm):
                nn.init.ones_(m.weight)


def build_biencoder(config=None, embedding_dim=None, dtype: str = "float32", **_):
    m = BiEncoder(dict(config) if config else {}, embedding_dim=embedding_dim)
    return m.to(getattr(torch, dtype))


class TrainRecipeForRetrieval(TrainFinetuneRecipeForNextTokenPrediction):
    def setup(self) -> None:
        self.cfg.model["_target_"] = "automodel_amd.recipes.llm.train_retrieval.build_biencoder"
        super().setup()
        self.temperature = self.cfg.get_by_dotted("retrieval.temperature", 0.05)

    def _forward_backward_step(self, batch, loss_scale):
        q = batch["query_ids"].to(self.device, non_blocking=True)
        p = batch["positive_ids"].to(self.device, non_blocking=True)
        loss = info_nce_loss(self.model.encode(q), self.model.encode(p),
                             temperature=self.temperature,
                             group=self.mesh.dp_group())
        (loss * q.shape[0] * loss_scale).backward()  # mean -> sum-comparable
        return (loss * q.shape[0]).detach()

    def _build_loader(self, dcfg: ConfigNode):
        from torch.utils.data import DataLoader, Dataset

        from automodel_amd.datasets.loader import StatefulLoader

        ds_cfg = dcfg.get("dataset", ConfigNode())
        if "_target_" in ds_cfg:
            dataset = ds_cfg.instantiate()
        else:
            class MockPairs(Dataset):
                def __init__(self, n=64, s=16, v=128):
                    self.n, self.s, self.v = n, s, v

                def __len__(self):
                    return self.n

                def __getitem__(self, i):
                    g = torch.Generator().manual_seed(i)
                    q = torch.randint(0, self.v, (self.s,), generator=g)
                    return {"query_ids": q, "positive_ids": q.roll(1)}

            dataset = MockPairs(ds_cfg.get("num_samples", 64), ds_cfg.get("seq_len", 16),
                                ds_cfg.get("vocab_size", 128))

        def collate(b):
            return {"query_ids": torch.stack([x["query_ids"] for x in b]),
                    "positive_ids": torch.stack([x["positive_ids"] for x in b])}

        return StatefulLoader(DataLoader(dataset, batch_size=dcfg.get("batch_size", 2),
                                         collate_fn=collate, drop_last=True))


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    cfg = load_yaml_config(argv[0])
    apply_overrides(cfg, parse_cli_overrides(argv[1:]))
    r = TrainRecipeForRetrieval(cfg)
    r.setup()
    r.run_train_validation_loop()


if __name__ == "__main__":
    main()
