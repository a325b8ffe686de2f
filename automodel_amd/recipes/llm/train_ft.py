"""Flagship LLM SFT / pretrain recipe (next-token prediction).

Reference behavior: nemo_automodel/recipes/llm/train_ft.py:436-1601
(TrainFinetuneRecipeForNextTokenPrediction: setup() builds mesh -> model ->
optimizer -> dataloaders -> schedulers; run_train_validation_loop() drives
grad-accum steps with token-count-normalized token-sum loss, grad clip,
metrics, checkpoint cadence).

MI355X layout: one process per GPU over RCCL; FSDP2 sharding with
reshard_after_forward=False (288 GB HBM3E); hot ops on in-tree HIP kernels.
"""

from __future__ import annotations

import os
import sys
import time
from typing import Any

import torch
import torch.distributed as dist

from automodel_amd.config.loader import ConfigNode, load_yaml_config, parse_cli_overrides, apply_overrides
from automodel_amd.datasets.loader import build_dataloader
from automodel_amd.datasets.mock import MockDataset, MockIterableDataset
from automodel_amd.loggers.metric_logger import MetricLogger, setup_logging
from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy
from automodel_amd.models.registry import build_model
from automodel_amd.optim.adamw import build_adamw
from automodel_amd.optim.lr_scheduler import WarmupDecayLR
from automodel_amd.parallel.fsdp import apply_fsdp
from automodel_amd.parallel.mesh import build_mesh, init_distributed
from automodel_amd.recipes.base import BaseRecipe
from automodel_amd.training.rng import StatefulRNG
from automodel_amd.training.step_scheduler import StepScheduler
from automodel_amd.training.utils import clip_grad_norm_, count_label_tokens, prepare_for_grad_accumulation


class TrainFinetuneRecipeForNextTokenPrediction(BaseRecipe):
    engine = None          # megatron_fsdp second engine (distributed.engine)

    def __init__(self, cfg: ConfigNode):
        super().__init__(cfg)
        self.logger = setup_logging()
        self.device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")

    # ------------------------------------------------------------------ setup
    def setup(self) -> None:
        cfg = self.cfg
        self.rank_id, self.local_rank, self.world = init_distributed()
        self.rng = StatefulRNG(seed=cfg.get("seed", 42), ranked=True)

        dist_cfg = cfg.get("distributed", ConfigNode())
        self.mesh = build_mesh(
            dp_replicate=dist_cfg.get("dp_replicate", 1),
            dp_shard=dist_cfg.get("dp_shard", -1),
            tp=dist_cfg.get("tp", 1),
            pp=dist_cfg.get("pp", 1),
            cp=dist_cfg.get("cp", 1),
            axis_timeouts=(dist_cfg.get("axis_timeouts")
                           and dict(dist_cfg.axis_timeouts.items())),
        )
        self.cp_size = self.mesh.dims.get("cp", 1)
        if self.cp_size > 1:
            from automodel_amd.parallel.cp import enable_cp
            enable_cp(self.mesh["cp"])

        # ---- model (meta init -> shard -> materialize/load)
        mcfg = cfg.model
        if "_target_" in mcfg:
            self.model = mcfg.instantiate()
        else:
            self.model = build_model(
                config=mcfg.get("config") and mcfg.config.to_dict(),
                pretrained_path=mcfg.get("pretrained_path"),
                architecture=mcfg.get("architecture"),
                backend=mcfg.get("backend") and mcfg.backend.to_dict(),
                dtype=mcfg.get("dtype", "bfloat16"),
            )

        # validate model capabilities against the requested mesh
        from automodel_amd.models.common.capabilities import validate_model_against_mesh
        problems = validate_model_against_mesh(self.model, self.mesh.dims)
        if problems:
            raise ValueError("model/mesh incompatibility: " + "; ".join(problems))

        # loss lives inside forward so FSDP keeps lm_head unsharded at use
        loss_cfg = cfg.get("loss_fn", ConfigNode())
        self.loss_fn = loss_cfg.maybe_instantiate(
            default=FusedLinearCrossEntropy(
                backend=loss_cfg.get("backend", "hip_fused"),
                chunk_size=loss_cfg.get("chunk_size", 2048),
            )
        )
        self.model.loss_fn = self.loss_fn

        # ---- fp8 linear swap (before sharding)
        q_cfg = cfg.get("quantization")
        if q_cfg and q_cfg.get("fp8", False):
            from automodel_amd.quantization.fp8 import apply_fp8_to_model
            n = apply_fp8_to_model(self.model)
            if self.rank == 0:
                self.logger.info(f"fp8: swapped {n} linears to Float8Linear")

        # ---- activation checkpointing (before sharding)
        ac_cfg = cfg.get("activation_checkpointing")
        if ac_cfg and ac_cfg.get("enabled", True):
            from automodel_amd.parallel.activation_checkpointing import apply_ac
            apply_ac(self.model, mode=ac_cfg.get("mode", "full"),
                     every_n=ac_cfg.get("every_n", 1))

        # ---- pipeline parallelism: split the model into this rank's stage
        self.pipeline = None
        if self.mesh.mesh is not None and self.mesh.pp_size > 1:
            from automodel_amd.parallel.pp import AutoPipeline, PipelineConfig
            pp_cfg = dist_cfg.get("pipeline", ConfigNode())
            if any(p.is_meta for p in self.model.parameters()):
                self.model.init_weights(device="cpu")
            self.pipeline = AutoPipeline(
                self.model, self.mesh["pp"],
                PipelineConfig(pp_size=self.mesh.pp_size,
                               schedule=pp_cfg.get("schedule", "1f1b"),
                               microbatches=pp_cfg.get("microbatches", self.mesh.pp_size),
                               virtual_stages=pp_cfg.get("virtual_stages", 1)),
                loss_fn=self.loss_fn, device=self.device,
            )
            self.model = self.pipeline.stage_module

        if self.mesh.mesh is not None and self.mesh.dims["tp"] > 1:
            from automodel_amd.parallel.tp import apply_tp
            apply_tp(self.model, self.mesh["tp"],
                     sequence_parallel=dist_cfg.get("sequence_parallel", False))
        # second sharding engine (reference megatron_fsdp.py:46): flat-shard
        # DP with jointly-sharded optimizer state; excludes FSDP2/PP/EP
        self._mfsdp = dist_cfg.get("engine") == "megatron_fsdp"
        if self._mfsdp:
            assert self.mesh.pp_size == 1 and self.mesh.dims.get("tp", 1) == 1 \
                and self.cp_size == 1, "megatron_fsdp engine is DP-only"
        if not self._mfsdp and self.mesh.mesh is not None and (
            self.mesh.dims["dp_shard"] > 1 or self.cp_size > 1
        ):
            fsdp_axis = "dp_shard_cp" if self.cp_size > 1 else "dp_shard"
            if self.mesh.dims.get("dp_replicate", 1) > 1:
                fsdp_mesh = self.mesh.mesh[("dp_replicate", fsdp_axis)]  # HSDP
            else:
                fsdp_mesh = self.mesh[fsdp_axis]
            apply_fsdp(
                self.model,
                fsdp_mesh,
                reshard_after_forward=dist_cfg.get("reshard_after_forward", False),
            )

        pretrained = mcfg.get("pretrained_path")
        if pretrained and os.path.exists(os.path.join(pretrained, "model.safetensors.index.json")) or \
           pretrained and any(f.endswith(".safetensors") for f in os.listdir(pretrained)):
            from automodel_amd.checkpoint.hf_loader import load_hf_weights
            load_hf_weights(self.model, pretrained, device=self.device)
        elif self.pipeline is None:
            self.model.init_weights(device=self.device)
        # (PP stages were materialized + initialized before the split)

        # ---- PEFT
        peft_cfg = cfg.get("peft")
        if peft_cfg:
            from automodel_amd.peft.lora import apply_lora_to_linear_modules
            apply_lora_to_linear_modules(self.model, peft_cfg)

        # ---- QAT (reference train_ft.py:218 applies after PEFT/model build)
        qat_cfg = cfg.get("qat")
        self.qat_enabled = bool(qat_cfg and qat_cfg.get("enabled", True))
        if self.qat_enabled:
            from automodel_amd.quantization.qat import prepare_qat
            n_qat = prepare_qat(self.model, qat_cfg)
            if self.rank == 0:
                self.logger.info(f"QAT: fake-quantizing {n_qat} linear modules")

        # ---- optimizer / schedulers
        opt_cfg = cfg.get("optimizer", ConfigNode())
        self.engine = None
        if self._mfsdp:
            from automodel_amd.parallel.megatron_fsdp import MegatronFSDPEngine
            self.engine = MegatronFSDPEngine(
                self.model,
                lr=opt_cfg.get("lr", 2e-5),
                weight_decay=opt_cfg.get("weight_decay", 0.01),
                betas=tuple(opt_cfg.get("betas", (0.9, 0.999))),
            )
            self.optimizer = self.engine.optimizer
        else:
            self.optimizer = opt_cfg.maybe_instantiate(model=self.model) or build_adamw(
                self.model,
                lr=opt_cfg.get("lr", 2e-5),
                weight_decay=opt_cfg.get("weight_decay", 0.01),
                betas=tuple(opt_cfg.get("betas", (0.9, 0.999))),
            )
        sched_cfg = cfg.get("step_scheduler", ConfigNode())
        self.step_scheduler = StepScheduler(
            grad_acc_steps=sched_cfg.get("grad_acc_steps", 1),
            ckpt_every_steps=sched_cfg.get("ckpt_every_steps", 0),
            val_every_steps=sched_cfg.get("val_every_steps", 0),
            max_steps=sched_cfg.get("max_steps"),
            num_epochs=sched_cfg.get("num_epochs", 1),
        )
        lr_cfg = cfg.get("lr_scheduler", ConfigNode())
        self.lr_scheduler = WarmupDecayLR(
            self.optimizer,
            warmup_steps=lr_cfg.get("warmup_steps", 0),
            total_steps=lr_cfg.get("total_steps", sched_cfg.get("max_steps")),
            decay=lr_cfg.get("decay", "constant"),
            min_lr_ratio=lr_cfg.get("min_lr_ratio", 0.0),
        )

        # ---- data
        self.train_loader = self._build_loader(cfg.get("dataloader", ConfigNode()))
        self.step_scheduler.dataloader = self.train_loader
        self.val_loader = None
        if cfg.get("validation") and cfg.validation.get("dataloader"):
            self.val_loader = self._build_loader(cfg.validation.dataloader)

        # ---- checkpointing
        ckpt_cfg = cfg.get("checkpoint")
        if ckpt_cfg and ckpt_cfg.get("enabled", True) and ckpt_cfg.get("checkpoint_dir"):
            from automodel_amd.checkpoint.checkpointing import Checkpointer
            self.checkpointer = Checkpointer(
                checkpoint_dir=ckpt_cfg.checkpoint_dir,
                model_save_format=ckpt_cfg.get("model_save_format", "safetensors"),
                save_consolidated=ckpt_cfg.get("save_consolidated", False),
                keep_last_n=ckpt_cfg.get("keep_last_n"),
                keep_top_k=ckpt_cfg.get("keep_top_k"),
                metric_higher_is_better=ckpt_cfg.get("metric_higher_is_better", False),
                async_save=ckpt_cfg.get("async_save", False),
            )

        # ---- EMA / NEFTune / stepped GC (training extras)
        self.ema = None
        if cfg.get("ema") and cfg.ema.get("enabled", True):
            from automodel_amd.training.extras import EMA
            self.ema = EMA(self.model, decay=cfg.ema.get("decay", 0.999))
        if cfg.get("neftune") and cfg.neftune.get("enabled", True):
            from automodel_amd.training.extras import apply_neftune
            emb = getattr(getattr(self.model, "model", None), "embed_tokens", None)
            if emb is not None:
                apply_neftune(emb, alpha=cfg.neftune.get("alpha", 5.0))
        self.gc = None
        gc_cfg = cfg.get("garbage_collection")
        if gc_cfg and gc_cfg.get("enabled", True):
            from automodel_amd.training.extras import SteppedGarbageCollector
            self.gc = SteppedGarbageCollector(gc_cfg.get("every_steps", 100))

        # ---- metric logging
        out_dir = cfg.get("output_dir", "outputs")
        self.metrics = MetricLogger(os.path.join(out_dir, "training.jsonl"))
        self.max_grad_norm = cfg.get("max_grad_norm", 1.0)

        restore = cfg.get("restore_from")
        if restore:
            self.load_checkpoint(restore)

    def _count_batch_tokens(self, batch: dict) -> torch.Tensor:
        if "labels" in batch:
            return count_label_tokens(batch["labels"])
        for k in ("input_ids", "query_ids"):
            if k in batch:
                return torch.tensor(batch[k].numel())
        return torch.tensor(1)

    def _dp_cp_group(self):
        if self.mesh.mesh is None:
            return None
        if self.cp_size > 1:
            return self.mesh.mesh["dp_cp"].get_group()
        return self.mesh.dp_group()

    def _build_loader(self, dcfg: ConfigNode):
        ds_cfg = dcfg.get("dataset", ConfigNode())
        if "_target_" in ds_cfg:
            dataset = ds_cfg.instantiate()
        else:
            kind = ds_cfg.get("kind", "mock_iterable")
            kwargs = {k: v for k, v in ds_cfg.items() if k != "kind"}
            dataset = (MockIterableDataset if kind == "mock_iterable" else MockDataset)(**kwargs)
        return build_dataloader(
            dataset,
            batch_size=dcfg.get("batch_size", 1),
            shuffle=dcfg.get("shuffle", True),
            num_workers=dcfg.get("num_workers", 0),
            pad_token_id=dcfg.get("pad_token_id", 0),
            dp_rank=self.mesh.dp_rank if self.mesh.mesh is not None else 0,
            dp_world=self.mesh.dp_size,
            seed=self.cfg.get("seed", 42),
        )

    # ------------------------------------------------------------- train step
    def _pp_step(self, batches: list[dict], loss_scale: float) -> torch.Tensor:
        """One optimizer step through the pipeline schedule (pp > 1).
        The schedule splits the batch into microbatches; the loss is computed
        on the last stage and broadcast for metrics (reference
        train_ft.py:1188 PP loss broadcast)."""
        total = torch.zeros((), dtype=torch.float32, device=self.device)
        # grad accumulation: each schedule.step accumulates into .grad
        for batch in batches:
            input_ids = batch["input_ids"].to(self.device, non_blocking=True)
            labels = batch["labels"].to(self.device, non_blocking=True)
            losses = self.pipeline.step(input_ids=input_ids, target=labels)
            if self.pipeline.is_last and losses:
                total = total + sum(l.float() for l in losses)
        # scale grads: schedule backwards sum-of-microbatch losses; match the
        # non-PP loss_scale by scaling grads post-hoc
        for p in self.model.parameters():
            if p.grad is not None:
                p.grad.mul_(loss_scale)
        src = dist.get_process_group_ranks(self.mesh["pp"].get_group())[-1]
        dist.broadcast(total, src=src, group=self.mesh["pp"].get_group())
        return total

    def _forward_backward_step(self, batch: dict, loss_scale: float) -> torch.Tensor:
        if self.cp_size > 1:
            from automodel_amd.parallel.cp import shard_batch_cp
            cp_rank = self.mesh["cp"].get_local_rank()
            batch = shard_batch_cp(batch, cp_rank, self.cp_size)
        input_ids = batch["input_ids"].to(self.device, non_blocking=True)
        labels = batch["labels"].to(self.device, non_blocking=True)
        position_ids = batch.get("position_ids")
        if position_ids is not None:
            position_ids = position_ids.to(self.device, non_blocking=True)
        cu = batch.get("cu_seqlens")
        if cu is not None:
            from automodel_amd.ops.attention import set_varlen_context
            set_varlen_context(cu.to(self.device))
        try:
            loss = self.model(input_ids, labels=labels, position_ids=position_ids)
            (loss * loss_scale).backward()
        finally:
            if cu is not None:
                set_varlen_context(None)
        return loss.detach()

    def _run_train_optim_step(self, batches: list[dict]) -> dict[str, Any]:
        t0 = time.perf_counter()
        device = self.device
        num_label_tokens = torch.zeros((), dtype=torch.long)
        for b in batches:
            num_label_tokens += self._count_batch_tokens(b)
        num_label_tokens = num_label_tokens.to(device)
        if self.world > 1:
            # counted on the UNSHARDED batch (cp ranks share it): reduce over dp
            dist.all_reduce(num_label_tokens, group=self.mesh.dp_group())
        global_tokens = max(1, int(num_label_tokens.item()))

        # loss_sum / global_tokens * dp_cp_world compensates FSDP's mean-reduce
        # (with CP, each rank sees a token subset and FSDP reduces over
        # dp_shard_cp — reference train_ft.py:1186 scales by dp_cp_size);
        # the megatron_fsdp engine reduce-scatters with SUM, so no dp factor
        loss_scale = ((1.0 if self.engine is not None else self.mesh.dp_cp_size)
                      / global_tokens)
        total_loss = torch.zeros((), dtype=torch.float32, device=device)
        if self.pipeline is not None:
            total_loss = self._pp_step(batches, loss_scale).float()
        else:
            for i, batch in enumerate(batches):
                prepare_for_grad_accumulation(self.model, is_final_microbatch=(i == len(batches) - 1))
                total_loss += self._forward_backward_step(batch, loss_scale).float()
                if self.engine is not None:
                    self.engine.reduce_grads()

        if self.engine is not None:
            grad_norm = self.engine.clip_grad_norm(self.max_grad_norm)
        else:
            grad_norm = clip_grad_norm_(self.model.parameters(), self.max_grad_norm)
        if hasattr(self, "checkpointer") and hasattr(self.checkpointer, "maybe_wait_for_staging"):
            self.checkpointer.maybe_wait_for_staging()
        if self.engine is not None:
            self.engine.step()
        else:
            self.optimizer.step()
            self.optimizer.zero_grad(set_to_none=True)
        self.lr_scheduler.step()
        # MoE: aux-free gate-bias update + load metrics (reference
        # train_ft.py update_moe_gate_bias / load_balance_metrics)
        if self.ema is not None:
            self.ema.update(self.model)
        if self.gc is not None:
            self.gc.maybe_collect(self.step_scheduler.step)
        moe_metrics = {}
        if hasattr(self.model, "update_moe_gate_bias"):
            self.model.update_moe_gate_bias()
            from automodel_amd.moe.load_balance_metrics import load_balance_metrics
            moe_metrics = load_balance_metrics(self.model, group=self.mesh.dp_group())

        if self.world > 1:
            dist.all_reduce(total_loss, group=self._dp_cp_group())
        step_time = time.perf_counter() - t0
        ntok = int(num_label_tokens.item())
        return {
            "step": self.step_scheduler.step,
            **moe_metrics,
            "loss": total_loss.item() / max(1, ntok),
            "grad_norm": float(grad_norm),
            "lr": self.lr_scheduler.get_last_lr()[0],
            "num_label_tokens": ntok,
            "step_time_s": step_time,
            "tps": ntok / step_time,
            "mem_gb": (torch.cuda.max_memory_allocated() / 2**30) if torch.cuda.is_available() else 0.0,
        }

    @torch.no_grad()
    def _run_validation_epoch(self) -> dict[str, float]:
        self.model.eval()
        total_loss, total_tok = 0.0, 0
        for batch in self.val_loader:
            input_ids = batch["input_ids"].to(self.device, non_blocking=True)
            labels = batch["labels"].to(self.device, non_blocking=True)
            loss = self.model(input_ids, labels=labels)
            total_loss += float(loss)
            total_tok += int(count_label_tokens(labels))
        self.model.train()
        t = torch.tensor([total_loss, float(total_tok)], device=self.device)
        if self.world > 1:
            dist.all_reduce(t, group=self.mesh.dp_group())
        return {"val_loss": float(t[0] / max(1.0, float(t[1])))}

    # -------------------------------------------------------------- main loop
    def run_train_validation_loop(self) -> None:
        self.model.train()
        for epoch in self.step_scheduler.epochs:
            self.train_loader.set_epoch(epoch)
            for batches in self.step_scheduler:
                if getattr(self, "qat_enabled", False):
                    from automodel_amd.quantization.qat import maybe_enable_delayed_fake_quant
                    maybe_enable_delayed_fake_quant(self.model, self.step_scheduler.step)
                metrics = self._run_train_optim_step(batches)
                self.metrics.log(metrics)
                if self.rank == 0:
                    self.logger.info(
                        f"step {metrics['step']} | loss {metrics['loss']:.4f} | "
                        f"gnorm {metrics['grad_norm']:.3f} | lr {metrics['lr']:.2e} | "
                        f"tok {metrics['num_label_tokens']} | {metrics['step_time_s']*1e3:.0f} ms"
                    )
                if self.val_loader is not None and self.step_scheduler.is_val_step:
                    vm = self._run_validation_epoch()
                    self.metrics.log({"step": metrics["step"], **vm})
                if self.step_scheduler.is_ckpt_step and hasattr(self, "checkpointer"):
                    self.save_checkpoint(
                        os.path.join(self.checkpointer.checkpoint_dir,
                                     f"step_{self.step_scheduler.step}")
                    )
        self.metrics.close()


def main(argv: list[str] | None = None) -> None:
    argv = argv if argv is not None else sys.argv[1:]
    assert argv, "usage: python -m automodel_amd.recipes.llm.train_ft <cfg.yaml> [--a.b=c ...]"
    cfg = load_yaml_config(argv[0])
    apply_overrides(cfg, parse_cli_overrides(argv[1:]))
    recipe = TrainFinetuneRecipeForNextTokenPrediction(cfg)
    recipe.setup()
    recipe.run_train_validation_loop()
    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
