"""VLM SFT recipe (vision tower + LLM), config #3.

Reference behavior: nemo_automodel/recipes/vlm/finetune.py:397
(FinetuneRecipeForVLM: frozen-tower handling, VLM collators, same loop
structure as the LLM recipe). Reuses the LLM recipe loop; overrides model
construction and batch handling for pixel_values.
"""

from __future__ import annotations

import sys

import torch

from automodel_amd.config.loader import ConfigNode, apply_overrides, load_yaml_config, parse_cli_overrides
from automodel_amd.datasets.vlm.mock import MockVLMDataset, vlm_collate
from automodel_amd.recipes.llm.train_ft import TrainFinetuneRecipeForNextTokenPrediction


class FinetuneRecipeForVLM(TrainFinetuneRecipeForNextTokenPrediction):
    def setup(self) -> None:
        super().setup()
        freeze_cfg = self.cfg.get("freeze", ConfigNode())
        if freeze_cfg.get("vision_tower", False) and hasattr(self.model, "freeze_vision_tower"):
            self.model.freeze_vision_tower()

    def _forward_backward_step(self, batch: dict, loss_scale: float) -> torch.Tensor:
        input_ids = batch["input_ids"].to(self.device, non_blocking=True)
        labels = batch["labels"].to(self.device, non_blocking=True)
        pixel_values = batch.get("pixel_values")
        if pixel_values is not None:
            pixel_values = pixel_values.to(self.device, non_blocking=True)
        loss = self.model(input_ids, pixel_values=pixel_values, labels=labels)
        (loss * loss_scale).backward()
        return loss.detach()

    def _build_loader(self, dcfg: ConfigNode):
        ds_cfg = dcfg.get("dataset", ConfigNode())
        if "_target_" in ds_cfg:
            dataset = ds_cfg.instantiate()
        else:
            kwargs = {k: v for k, v in ds_cfg.items() if k != "kind"}
            dataset = MockVLMDataset(**kwargs)
        from torch.utils.data import DataLoader

        from automodel_amd.datasets.loader import StatefulLoader

        loader = DataLoader(
            dataset,
            batch_size=dcfg.get("batch_size", 1),
            shuffle=False,
            collate_fn=vlm_collate,
            drop_last=True,
        )
        return StatefulLoader(loader)


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    cfg = load_yaml_config(argv[0])
    apply_overrides(cfg, parse_cli_overrides(argv[1:]))
    r = FinetuneRecipeForVLM(cfg)
    r.setup()
    r.run_train_validation_loop()


if __name__ == "__main__":
    main()
