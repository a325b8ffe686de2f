"""Discrete-diffusion LM (MDLM) SFT recipe.

Reference behavior: nemo_automodel/recipes/dllm/train_ft.py (corrupt the
batch with the mask token at a sampled noise level, run the model
BIDIRECTIONALLY, apply the 1/t-weighted masked CE). Reuses the flagship
recipe's setup (mesh/model/optimizer/data/checkpoint); only the per-batch
forward/backward is replaced, and the model is built with
``bidirectional: true`` (denoising needs full context).
"""

from __future__ import annotations

import sys
from typing import Any

import torch

from automodel_amd.config.loader import ConfigNode, apply_overrides, load_yaml_config, parse_cli_overrides
from automodel_amd.loss.dllm import MDLMCrossEntropyLoss, mdlm_corrupt
from automodel_amd.recipes.llm.train_ft import TrainFinetuneRecipeForNextTokenPrediction


class TrainDiffusionLMRecipe(TrainFinetuneRecipeForNextTokenPrediction):
    def setup(self) -> None:
        # denoising attends in both directions
        self.cfg.set_by_dotted("model.config.bidirectional", True)
        super().setup()
        d = self.cfg.get("dllm", ConfigNode())
        self.mask_token_id = d.get("mask_token_id",
                                   self.model.config.vocab_size - 1)
        self.noise_eps = d.get("noise_eps", 1e-3)
        self.dllm_loss = MDLMCrossEntropyLoss()

    def _forward_backward_step(self, batch: dict, loss_scale: float) -> torch.Tensor:
        ids = batch["input_ids"].to(self.device, non_blocking=True)
        loss_mask = (batch["labels"].to(self.device) != -100) \
            if "labels" in batch else torch.ones_like(ids, dtype=torch.bool)
        noisy, noise_mask, p_mask = mdlm_corrupt(
            ids, self.mask_token_id, loss_mask, self.noise_eps)
        logits = self.model(noisy)
        n_tok = int(loss_mask.sum())
        loss = self.dllm_loss(logits, ids, noise_mask, p_mask, loss_mask,
                              num_diffusion_tokens=max(n_tok, 1))
        (loss * loss_scale).backward()
        return loss.detach()


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    cfg = load_yaml_config(argv[0])
    apply_overrides(cfg, parse_cli_overrides(argv[1:]))
    r = TrainDiffusionLMRecipe(cfg)
    r.setup()
    r.run_train_validation_loop()


if __name__ == "__main__":
    main()
