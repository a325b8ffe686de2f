"""Pipeline parallelism: model splitting + torch pipelining schedules.

Reference behavior: nemo_automodel/components/distributed/pipelining/
autopipeline.py:52 (AutoPipeline: split an HF-style model by FQN into stages,
build PipelineStage objects, select a schedule) and functional.py:597
(split_model_into_stages). Here the split is structural: stage 0 keeps
embed_tokens, every stage keeps a contiguous slice of decoder layers, the last
stage keeps norm + lm_head and computes the loss.

P2P activations ride RCCL send/recv over xGMI.
"""

from __future__ import annotations

import copy
from dataclasses import dataclass

import torch
import torch.nn as nn
from torch.distributed.device_mesh import DeviceMesh
from torch.distributed.pipelining import PipelineStage, Schedule1F1B, ScheduleGPipe
from torch.distributed.pipelining.schedules import ScheduleInterleaved1F1B


@dataclass
class PipelineConfig:
    pp_size: int = 1
    schedule: str = "1f1b"          # 1f1b | gpipe | interleaved_1f1b
    microbatches: int = 1


def split_layer_ranges(num_layers: int, num_stages: int) -> list[tuple[int, int]]:
    """Contiguous near-even split (first stages get the remainder)."""
    base, rem = divmod(num_layers, num_stages)
    ranges = []
    start = 0
    for i in range(num_stages):
        n = base + (1 if i < rem else 0)
        ranges.append((start, start + n))
        start += n
    return ranges


class CausalLMStage(nn.Module):
    """One pipeline stage of a Llama-style causal LM.

    forward(x) where x is input_ids (stage 0) or hidden states; the last
    stage returns hidden states (loss handled by the schedule's loss_fn so
    fused linear CE sees lm_head.weight).
    """

    def __init__(self, full_model: nn.Module, stage_idx: int, num_stages: int):
        super().__init__()
        cfg = full_model.config
        self.stage_idx = stage_idx
        self.num_stages = num_stages
        self.is_first = stage_idx == 0
        self.is_last = stage_idx == num_stages - 1
        lo, hi = split_layer_ranges(cfg.num_hidden_layers, num_stages)[stage_idx]
        self.layer_range = (lo, hi)

        self.embed_tokens = full_model.model.embed_tokens if self.is_first else None
        self.layers = nn.ModuleList(full_model.model.layers[lo:hi])
        self.norm = full_model.model.norm if self.is_last else None
        self.lm_head = full_model.lm_head if self.is_last else None
        # rope tables are cheap; every stage keeps its own copy
        self.register_buffer("rope_cos", full_model.model.rope_cos, persistent=False)
        self.register_buffer("rope_sin", full_model.model.rope_sin, persistent=False)
        self.loss_fn = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.is_first:
            x = self.embed_tokens(x)
        S = x.shape[1]
        cos, sin = self.rope_cos[:S], self.rope_sin[:S]
        if cos.dtype != torch.float32:
            cos, sin = cos.float(), sin.float()
        for layer in self.layers:
            x = layer(x, cos, sin)
        if self.is_last:
            x = self.norm(x)
        return x


class AutoPipeline:
    """Builds stages + schedule for this rank (reference autopipeline.py:132)."""

    def __init__(
        self,
        model: nn.Module,
        pp_mesh: DeviceMesh,
        config: PipelineConfig,
        loss_fn=None,
        device: torch.device | str = "cpu",
    ):
        self.pp_size = pp_mesh.size()
        self.pp_rank = pp_mesh.get_local_rank()
        self.group = pp_mesh.get_group()
        self.config = config
        stage_mod = CausalLMStage(model, self.pp_rank, self.pp_size)
        stage_mod.loss_fn = loss_fn
        device = torch.device(device)
        if any(p.is_meta for p in stage_mod.parameters()):
            stage_mod.to_empty(device=device)
        else:
            stage_mod.to(device)
        self.stage_module = stage_mod
        self.stage = PipelineStage(
            stage_mod, self.pp_rank, self.pp_size, device, group=self.group
        )

        def schedule_loss(output, target):
            # last-stage hidden -> fused linear CE against lm_head weight
            return stage_mod.loss_fn(output, stage_mod.lm_head.weight, target)

        sched_cls = {
            "1f1b": Schedule1F1B,
            "gpipe": ScheduleGPipe,
            "interleaved_1f1b": ScheduleInterleaved1F1B,
        }[config.schedule]
        n_mb = max(config.microbatches, self.pp_size)
        if sched_cls is ScheduleInterleaved1F1B:
            self.schedule = sched_cls([self.stage], n_microbatches=n_mb,
                                      loss_fn=schedule_loss if loss_fn else None)
        else:
            self.schedule = sched_cls(self.stage, n_microbatches=n_mb,
                                      loss_fn=schedule_loss if loss_fn else None)

    @property
    def is_first(self) -> bool:
        return self.pp_rank == 0

    @property
    def is_last(self) -> bool:
        return self.pp_rank == self.pp_size - 1

    def step(self, input_ids: torch.Tensor | None = None,
             target: torch.Tensor | None = None) -> list[torch.Tensor] | None:
        """Run one scheduled fwd+bwd over the microbatch split."""
        losses: list[torch.Tensor] = []
        if self.is_first:
            self.schedule.step(input_ids.contiguous(), target=target, losses=losses)
        elif self.is_last:
            self.schedule.step(target=target, losses=losses)
        else:
            self.schedule.step()
        return losses if self.is_last else None
