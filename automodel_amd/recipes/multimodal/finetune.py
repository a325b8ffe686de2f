"""Unified-multimodal (omni) finetune recipe: understanding CE + visual-
generation flow-matching MSE over packed mixed-modality batches.

Reference behavior: nemo_automodel/recipes/multimodal/finetune.py
(FinetuneRecipeForMultimodal: BAGEL-style packed batches; per-token CE
reduced as ``ce.sum() * world / total_ce_tokens`` and per-token MSE as
``mse.sum() * world / total_mse_tokens`` with token counts all-reduced
across ranks; AdamW(0.9, 0.95, eps=1e-15, wd=0); constant LR with warmup;
Stage 1 = understanding-only CE, Stage 2 = joint CE + flow MSE).

MI355X-native composition: models/omni OmniForUnifiedMultimodal (in-tree
vision tower + llama trunk + TinyVAE latents + flow head) on DP/FSDP2.

Run: python -m automodel_amd.recipes.multimodal.finetune cfg.yaml [--a.b=c]
"""

from __future__ import annotations

import sys

import torch
import torch.distributed as dist

from automodel_amd.config.loader import (
    ConfigNode,
    apply_overrides,
    load_yaml_config,
    parse_cli_overrides,
)
from automodel_amd.models.llama.model import LlamaConfig
from automodel_amd.models.omni.model import OmniConfig, OmniForUnifiedMultimodal
from automodel_amd.models.vlm.model import VisionConfig
from automodel_amd.parallel.mesh import build_mesh, init_distributed
from automodel_amd.recipes.base import BaseRecipe
from automodel_amd.training.rng import StatefulRNG
from automodel_amd.training.step_scheduler import StepScheduler


class MockOmniDataset(torch.utils.data.Dataset):
    """Packed mixed-modality samples: text + image-understanding slots +
    (stage 2) generation-latent slots."""

    def __init__(self, num_samples: int, seq_len: int, vocab_size: int,
                 image_size: int, patch_size: int, n_gen_tokens: int,
                 image_token_id: int = 3, gen_token_id: int = 4,
                 stage: int = 2, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.samples = []
        n_patches = (image_size // patch_size) ** 2
        for i in range(num_samples):
            ids = torch.randint(5, vocab_size, (seq_len,), generator=g)
            ids[1:1 + n_patches] = image_token_id
            labels = ids.clone()
            gen = None
            if stage >= 2:
                s0 = 2 + n_patches
                ids[s0:s0 + n_gen_tokens] = gen_token_id
                labels = ids.clone()
                gen = torch.randn(3, image_size, image_size, generator=g)
            labels[(ids == image_token_id) | (ids == gen_token_id)] = -100
            self.samples.append({
                "input_ids": ids,
                "labels": labels,
                "pixel_values": torch.randn(3, image_size, image_size, generator=g),
                "gen_images": gen,
            })

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, i):
        return self.samples[i]


def omni_collate(items):
    out = {
        "input_ids": torch.stack([it["input_ids"] for it in items]),
        "labels": torch.stack([it["labels"] for it in items]),
        "pixel_values": torch.stack([it["pixel_values"] for it in items]),
    }
    gens = [it["gen_images"] for it in items if it["gen_images"] is not None]
    out["gen_images"] = torch.stack(gens) if gens else None
    return out


class FinetuneRecipeForMultimodal(BaseRecipe):
    def setup(self) -> None:
        cfg = self.cfg
        self.rank_id, _, self.world = init_distributed()
        self.rng = StatefulRNG(seed=cfg.get("seed", 4396), ranked=True)
        self.device = (torch.device("cuda") if torch.cuda.is_available()
                       else torch.device("cpu"))
        self.mesh = build_mesh(dp_shard=-1)
        self.stage = cfg.get("stage", 2)

        mcfg = cfg.model.config
        text = LlamaConfig(**dict(mcfg.text.items()))
        vision = VisionConfig(**dict(mcfg.vision.items()))
        self.model = OmniForUnifiedMultimodal(OmniConfig(
            text=text, vision=vision,
            latent_dim=mcfg.get("latent_dim", 16),
            gen_patch=mcfg.get("gen_patch", 2)))
        self.model.init_weights(device=self.device)
        freeze = cfg.get("freeze", ConfigNode())
        if freeze.get("vision_tower", False):
            self.model.freeze_vision_tower()
        if self.world > 1:
            from automodel_amd.parallel.fsdp import apply_fsdp
            apply_fsdp(self.model, self.mesh["dp_shard"],
                       param_dtype=torch.float32, reduce_dtype=torch.float32)
        ocfg = cfg.get("optimizer", ConfigNode())
        self.optimizer = torch.optim.AdamW(
            [p for p in self.model.parameters() if p.requires_grad],
            lr=ocfg.get("lr", 2e-5), betas=(0.9, 0.95), eps=1e-15,
            weight_decay=0.0)
        self.base_lr = ocfg.get("lr", 2e-5)
        self.warmup_steps = ocfg.get("warmup_steps", 0)

        dcfg = cfg.dataloader
        ds_args = dict(dcfg.dataset.items())
        ds = MockOmniDataset(
            num_samples=ds_args.get("num_samples", 8),
            seq_len=ds_args.get("seq_len", 64),
            vocab_size=text.vocab_size,
            image_size=vision.image_size, patch_size=vision.patch_size,
            n_gen_tokens=ds_args.get("n_gen_tokens", 4),
            stage=self.stage, seed=ds_args.get("seed", 42))
        sampler = None
        if self.world > 1:
            sampler = torch.utils.data.DistributedSampler(
                ds, num_replicas=self.world, rank=self.rank_id)
        self.loader = torch.utils.data.DataLoader(
            ds, batch_size=dcfg.get("batch_size", 2), shuffle=False,
            sampler=sampler, collate_fn=omni_collate, drop_last=True)
        self.step_scheduler = StepScheduler(
            max_steps=cfg.get("step_scheduler", ConfigNode()).get("max_steps", 5),
            grad_acc_steps=cfg.get("step_scheduler", ConfigNode()).get("grad_acc_steps", 1),
            dataloader=self.loader)
        self.metrics: list[dict] = []

    def _apply_warmup(self, step: int) -> None:
        if self.warmup_steps and step < self.warmup_steps:
            lr = self.base_lr * (step + 1) / self.warmup_steps
            for gparam in self.optimizer.param_groups:
                gparam["lr"] = lr

    def _run_step(self, batches) -> dict:
        dev = self.device
        ce_sum = torch.zeros(1, device=dev)
        mse_sum = torch.zeros(1, device=dev)
        counts = torch.zeros(2, device=dev)          # [ce_tokens, mse_tokens]
        outs = []
        for batch in batches:
            out = self.model(
                batch["input_ids"].to(dev),
                pixel_values=batch["pixel_values"].to(dev),
                gen_images=(batch["gen_images"].to(dev)
                            if self.stage >= 2 and batch["gen_images"] is not None
                            else None),
                labels=batch["labels"].to(dev))
            counts[0] += out.get("ce_tokens", 0)
            counts[1] += out.get("mse_tokens", 0)
            outs.append(out)
        # reference reduction: sum-losses scaled by world / total tokens,
        # token totals all-reduced over dp
        if self.world > 1:
            dist.all_reduce(counts)
        ce_w = self.world / counts[0].clamp(min=1.0)
        mse_w = self.world / counts[1].clamp(min=1.0)
        for out in outs:
            loss = out.get("ce", 0.0) * ce_w
            if self.stage >= 2 and "mse" in out:
                loss = loss + out["mse"] * mse_w
            loss.backward()
            ce_sum += out.get("ce", torch.zeros(1, device=dev)).detach()
            if "mse" in out:
                mse_sum += out["mse"].detach()
        torch.nn.utils.clip_grad_norm_(self.model.parameters(), 1.0)
        self.optimizer.step()
        self.optimizer.zero_grad(set_to_none=True)
        return {"ce": float(ce_sum / counts[0].clamp(min=1.0)),
                "mse": float(mse_sum / counts[1].clamp(min=1.0)),
                "ce_tokens": int(counts[0]), "mse_tokens": int(counts[1])}

    def run_train_validation_loop(self) -> None:
        self.model.train()
        max_steps = self.cfg.get("step_scheduler", ConfigNode()).get("max_steps", 5)
        step = 0
        while step < max_steps:
            for batches in self.step_scheduler:
                self._apply_warmup(step)
                self.metrics.append(self._run_step(batches))
                step += 1
                if step >= max_steps:
                    break


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    cfg = load_yaml_config(argv[0])
    apply_overrides(cfg, parse_cli_overrides(argv[1:]))
    r = FinetuneRecipeForMultimodal(cfg)
    r.setup()
    r.run_train_validation_loop()


if __name__ == "__main__":
    main()
