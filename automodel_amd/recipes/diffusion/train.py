"""Image-diffusion training recipe: rectified flow matching on a DiT.

Reference behavior: nemo_automodel/recipes/diffusion/train.py (diffusion
training with flow-matching objectives; the reference defaults to pure
Ulysses CP for diffusion — out of scope here, DP/FSDP only).

Objective (rectified flow): x_t = (1-t) x0 + t eps, target v = eps - x0,
loss = ||model(x_t, t) - v||^2 averaged over pixels.

Run: python -m automodel_amd.recipes.diffusion.train cfg.yaml [--a.b=c]
"""

from __future__ import annotations

import sys

import torch

from automodel_amd.config.loader import ConfigNode, apply_overrides, load_yaml_config, parse_cli_overrides
from automodel_amd.models.dit.model import DiTConfig, DiTForFlowMatching
from automodel_amd.optim.adamw import build_adamw
from automodel_amd.parallel.mesh import build_mesh, init_distributed
from automodel_amd.recipes.base import BaseRecipe
from automodel_amd.training.rng import StatefulRNG
from automodel_amd.training.step_scheduler import StepScheduler


class MockImageDataset(torch.utils.data.Dataset):
    def __init__(self, num_samples: int, image_size: int, channels: int = 3, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.data = torch.randn(num_samples, channels, image_size, image_size,
                                generator=g)

    def __len__(self):
        return len(self.data)

    def __getitem__(self, i):
        return {"pixel_values": self.data[i]}


class TrainDiffusionRecipe(BaseRecipe):
    def setup(self) -> None:
        cfg = self.cfg
        self.rank_id, _, self.world = init_distributed()
        self.rng = StatefulRNG(seed=cfg.get("seed", 42), ranked=True)
        self.device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        self.mesh = build_mesh(dp_shard=-1)

        mcfg = dict(cfg.model.config.items())
        self.model = DiTForFlowMatching(DiTConfig(**mcfg))
        self.model.init_weights(device=self.device)
        if self.world > 1:
            from automodel_amd.parallel.fsdp import apply_fsdp
            apply_fsdp(self.model, self.mesh["dp_shard"],
                       param_dtype=torch.float32, reduce_dtype=torch.float32)
        self.optimizer = build_adamw(self.model, lr=cfg.get("optimizer", ConfigNode()).get("lr", 1e-4))

        dcfg = cfg.dataloader
        ds = MockImageDataset(dcfg.dataset.get("num_samples", 16),
                              mcfg.get("image_size", 32))
        self.loader = torch.utils.data.DataLoader(
            ds, batch_size=dcfg.get("batch_size", 4), shuffle=True)
        self.step_scheduler = StepScheduler(
            max_steps=cfg.get("step_scheduler", ConfigNode()).get("max_steps", 10),
            grad_acc_steps=1, dataloader=self.loader)
        self.losses: list[float] = []

    def _flow_loss(self, x0: torch.Tensor) -> torch.Tensor:
        B = x0.shape[0]
        t = torch.rand(B, device=x0.device)
        eps = torch.randn_like(x0)
        xt = (1 - t[:, None, None, None]) * x0 + t[:, None, None, None] * eps
        v = self.model(xt, t)
        return torch.nn.functional.mse_loss(v, eps - x0)

    def run_train_validation_loop(self) -> None:
        self.model.train()
        max_steps = self.cfg.get("step_scheduler", ConfigNode()).get("max_steps", 10)
        while len(self.losses) < max_steps:
            for batches in self.step_scheduler:
                for batch in batches:
                    x0 = batch["pixel_values"].to(self.device)
                    loss = self._flow_loss(x0)
                    loss.backward()
                self.optimizer.step()
                self.optimizer.zero_grad(set_to_none=True)
                self.losses.append(float(loss.detach()))
                if len(self.losses) >= max_steps:
                    break


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    cfg = load_yaml_config(argv[0])
    apply_overrides(cfg, parse_cli_overrides(argv[1:]))
    r = TrainDiffusionRecipe(cfg)
    r.setup()
    r.run_train_validation_loop()


if __name__ == "__main__":
    main()
