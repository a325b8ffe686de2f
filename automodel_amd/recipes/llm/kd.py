"""Knowledge-distillation recipe: frozen teacher + student.

Reference behavior: nemo_automodel/recipes/llm/kd.py (teacher/student KD with
forward-KL + CE mixture). Reuses the finetune loop; builds a second (frozen)
teacher model and overrides the loss step.
"""

from __future__ import annotations

import sys

import torch

from automodel_amd.config.loader import ConfigNode, apply_overrides, load_yaml_config, parse_cli_overrides
from automodel_amd.loss.kd_loss import KDLoss
from automodel_amd.models.registry import build_model
from automodel_amd.recipes.llm.train_ft import TrainFinetuneRecipeForNextTokenPrediction


class KDRecipeForNextTokenPrediction(TrainFinetuneRecipeForNextTokenPrediction):
    def setup(self) -> None:
        super().setup()
        tcfg = self.cfg.teacher
        self.teacher = build_model(
            config=tcfg.get("config") and tcfg.config.to_dict(),
            pretrained_path=tcfg.get("pretrained_path"),
            architecture=tcfg.get("architecture"),
            dtype=tcfg.get("dtype", "bfloat16"),
        )
        if tcfg.get("pretrained_path"):
            from automodel_amd.checkpoint.hf_loader import load_hf_weights
            load_hf_weights(self.teacher, tcfg.pretrained_path, device=self.device)
        else:
            self.teacher.init_weights(device=self.device)
        self.teacher.eval()
        for p in self.teacher.parameters():
            p.requires_grad_(False)
        kd_cfg = self.cfg.get("kd", ConfigNode())
        self.kd_loss = KDLoss(
            alpha=kd_cfg.get("alpha", 0.5),
            temperature=kd_cfg.get("temperature", 1.0),
            chunk_size=kd_cfg.get("chunk_size", 2048),
        )

    def _forward_backward_step(self, batch: dict, loss_scale: float) -> torch.Tensor:
        input_ids = batch["input_ids"].to(self.device, non_blocking=True)
        labels = batch["labels"].to(self.device, non_blocking=True)
        with torch.no_grad():
            teacher_logits = self.teacher(input_ids)
        student_logits = self.model(input_ids)
        loss = self.kd_loss(student_logits, teacher_logits, labels)
        (loss * loss_scale).backward()
        return loss.detach()


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    cfg = load_yaml_config(argv[0])
    apply_overrides(cfg, parse_cli_overrides(argv[1:]))
    r = KDRecipeForNextTokenPrediction(cfg)
    r.setup()
    r.run_train_validation_loop()


if __name__ == "__main__":
    main()
