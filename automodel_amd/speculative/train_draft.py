"""EAGLE draft training recipe: distill the frozen target into the draft.

Reference behavior: nemo_automodel/components/speculative/eagle/
peagle_trainer.py + core.py TTT (frozen target provides aux hidden states
and next-token distributions; the draft learns to predict the target's
choice one step ahead from [embed(token_t), carry_{t-1}]). The loss here is
the standard EAGLE pair: soft CE against the target's next-token
distribution + optional hard CE on the data labels.

Alignment (draft predicts one step further than the target position that
produced its carry): with tokens x[0..S-1], target aux-carry h[0..S-1] and
target logits z[0..S-1] (z_t is the distribution of x_{t+1}):

  draft input tokens  x[1..S-1]
  draft input carry   h[0..S-2]
  draft target dist   z[1..S-1]   (i.e. predict x_{t+2} at slot t+1)
"""

from __future__ import annotations

import sys
import time
from typing import Any

import torch
import torch.nn.functional as F

from automodel_amd.config.loader import ConfigNode, apply_overrides, load_yaml_config, parse_cli_overrides
from automodel_amd.speculative.draft import EagleDraftConfig, EagleDraftModel


def eagle_distill_loss(draft_logits: torch.Tensor, target_logits: torch.Tensor,
                       hard_labels: torch.Tensor | None = None,
                       hard_weight: float = 0.1) -> torch.Tensor:
    """Soft CE vs target distribution (+ optional hard CE on data labels)."""
    t = target_logits.float().log_softmax(-1)
    d = draft_logits.float().log_softmax(-1)
    soft = -(t.exp() * d).sum(-1).mean()
    if hard_labels is None or hard_weight == 0.0:
        return soft
    hard = F.cross_entropy(draft_logits.reshape(-1, draft_logits.shape[-1]).float(),
                           hard_labels.reshape(-1), ignore_index=-100)
    return soft + hard_weight * hard


class TrainEagleDraftRecipe:
    """Minimal trainer: frozen target + trainable draft on mock/instruction
    data. Shares the step/metric conventions of the main finetune recipe."""

    def __init__(self, cfg: ConfigNode):
        self.cfg = cfg

    def setup(self) -> None:
        cfg = self.cfg
        torch.manual_seed(cfg.get("seed", 42))
        from automodel_amd.models.registry import build_model

        device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = device
        self.target = build_model(
            config=cfg.model.config.to_dict(),
            architecture=cfg.model.get("architecture"),
            dtype=cfg.model.get("dtype", "float32"),
            meta_init=False,
            device=device,
        )
        pretrained = cfg.model.get("pretrained_path")
        if pretrained:
            from automodel_amd.checkpoint.hf_loader import load_hf_weights

            load_hf_weights(self.target, pretrained, device=device)
        self.target.eval()
        for p in self.target.parameters():
            p.requires_grad_(False)

        draft_over = dict(cfg.get("draft", ConfigNode()).items())
        tie = draft_over.pop("tie_embeddings", True)
        dcfg = EagleDraftConfig.from_target(self.target.config, **draft_over)
        self.draft = EagleDraftModel(dcfg).to(device)
        if tie:
            self.draft.tie_to_target(self.target)

        opt_cfg = cfg.get("optimizer", ConfigNode())
        self.optimizer = torch.optim.AdamW(
            [p for p in self.draft.parameters() if p.requires_grad],
            lr=opt_cfg.get("lr", 1e-4), weight_decay=opt_cfg.get("weight_decay", 0.01),
        )
        self.hard_weight = cfg.get("hard_label_weight", 0.1)
        # EAGLE-3 training-time multi-step unroll (TTT): after the first
        # step the draft conditions on its OWN previous hidden state, so
        # training sees the same drift it will face at decode time
        # (reference train_eagle3.py recipe_args.ttt_steps)
        self.ttt_steps = cfg.get("ttt_steps", 1)
        self.ttt_decay = cfg.get("ttt_decay", 1.0)
        self.max_steps = cfg.get_by_dotted("step_scheduler.max_steps", 100)

        dl_cfg = cfg.get("dataloader", ConfigNode())
        ds = dl_cfg.get("dataset", ConfigNode())
        from automodel_amd.datasets.mock import MockDataset

        self.seq_len = ds.get("seq_len", 128)
        self.batch_size = dl_cfg.get("batch_size", 2)
        self.dataset = MockDataset(
            num_samples=ds.get("num_samples", 256), seq_len=self.seq_len,
            vocab_size=self.target.config.vocab_size, seed=cfg.get("seed", 42),
        )

    def train_step(self, input_ids: torch.Tensor) -> dict[str, Any]:
        with torch.no_grad():
            z, aux = self.target.forward_with_aux(input_ids)
        # fuse_aux is the draft's trainable fc — OUTSIDE no_grad
        carry = self.draft.fuse_aux([a.detach() for a in aux])[:, :-1]
        ids, tgt, hard = input_ids[:, 1:], z[:, 1:], input_ids[:, 1:]
        total = None
        accs = []
        for s in range(self.ttt_steps):
            h = self.draft.backbone(ids, carry)
            d_logits = self.draft.lm_head(h)
            step_loss = eagle_distill_loss(d_logits, tgt, hard, self.hard_weight)
            w = self.ttt_decay ** s
            total = step_loss * w if total is None else total + step_loss * w
            # per-unroll-depth draft accuracy vs the target's greedy choice
            # (the number that predicts decode acceptance at depth s+1)
            with torch.no_grad():
                accs.append(float((d_logits.argmax(-1) == tgt.argmax(-1))
                                  .float().mean()))
            if s + 1 < self.ttt_steps:
                # next depth: the draft rides its OWN hidden, shifted one
                carry = h[:, :-1]
                ids, tgt, hard = ids[:, 1:], tgt[:, 1:], hard[:, 1:]
        self.optimizer.zero_grad(set_to_none=True)
        total.backward()
        self.optimizer.step()
        out = {"loss": float(total.detach()), "draft_top1_agreement": accs[0]}
        for s, a in enumerate(accs[1:], start=2):
            out[f"draft_top1_agreement_depth{s}"] = a
        return out

    def run_train_validation_loop(self) -> list[dict]:
        """CLI/launcher entry point (same contract as the other recipes)."""
        return self.run()

    def run(self) -> list[dict]:
        logs = []
        step = 0
        while step < self.max_steps:
            for i in range(0, len(self.dataset), self.batch_size):
                if step >= self.max_steps:
                    break
                batch = [self.dataset[j]["input_ids"]
                         for j in range(i, min(i + self.batch_size, len(self.dataset)))]
                ids = torch.stack(batch).to(self.device)
                t0 = time.perf_counter()
                m = self.train_step(ids)
                m["step"] = step
                m["step_time_s"] = time.perf_counter() - t0
                logs.append(m)
                step += 1
        return logs


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    cfg = load_yaml_config(argv[0])
    apply_overrides(cfg, parse_cli_overrides(argv[1:]))
    r = TrainEagleDraftRecipe(cfg)
    r.setup()
    logs = r.run()
    print(logs[-1] if logs else "no steps")


if __name__ == "__main__":
    main()
