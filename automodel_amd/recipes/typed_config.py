"""Typed view over the recipe YAML sections.

Reference behavior: nemo_automodel/recipes/_typed_config.py:15-41
(RecipeConfig: typed façade coercing YAML sections into component config
dataclasses with .build()). Only this module knows the YAML schema — the
components stay YAML-agnostic (reference invariant).
"""

from __future__ import annotations

from dataclasses import dataclass, field

from automodel_amd.config.loader import ConfigNode


@dataclass
class DistributedConfig:
    dp_replicate: int = 1
    dp_shard: int = -1
    tp: int = 1
    pp: int = 1
    cp: int = 1
    sequence_parallel: bool = False
    reshard_after_forward: bool = False
    pipeline: dict = field(default_factory=dict)


@dataclass
class OptimizerConfig:
    lr: float = 2e-5
    weight_decay: float = 0.01
    betas: tuple = (0.9, 0.999)
    eps: float = 1e-8
    fused: bool = True

    def build(self, model):
        from automodel_amd.optim.adamw import build_adamw

        return build_adamw(model, lr=self.lr, betas=tuple(self.betas), eps=self.eps,
                           weight_decay=self.weight_decay, fused=self.fused)


@dataclass
class StepSchedulerConfig:
    grad_acc_steps: int = 1
    ckpt_every_steps: int = 0
    val_every_steps: int = 0
    max_steps: int | None = None
    num_epochs: int = 1

    def build(self):
        from automodel_amd.training.step_scheduler import StepScheduler

        return StepScheduler(**self.__dict__)


@dataclass
class CheckpointConfig:
    enabled: bool = True
    checkpoint_dir: str | None = None
    model_save_format: str = "safetensors"
    save_consolidated: bool = False
    keep_last_n: int | None = None
    async_save: bool = False

    def build(self):
        from automodel_amd.checkpoint.checkpointing import Checkpointer

        kw = {k: v for k, v in self.__dict__.items() if k != "enabled"}
        return Checkpointer(**kw)


@dataclass
class LossConfig:
    backend: str = "hybrid"
    chunk_size: int = 4096

    def build(self):
        from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy

        return FusedLinearCrossEntropy(backend=self.backend, chunk_size=self.chunk_size)


def _coerce(cls, node: ConfigNode | dict | None):
    if node is None:
        return cls()
    d = node.to_dict() if isinstance(node, ConfigNode) else dict(node)
    known = {f for f in cls.__dataclass_fields__}
    return cls(**{k: v for k, v in d.items() if k in known})


class RecipeConfig:
    """Typed accessor over a raw ConfigNode (sections coerced lazily)."""

    def __init__(self, cfg: ConfigNode):
        self.raw = cfg

    @property
    def distributed(self) -> DistributedConfig:
        return _coerce(DistributedConfig, self.raw.get("distributed"))

    @property
    def optimizer(self) -> OptimizerConfig:
        return _coerce(OptimizerConfig, self.raw.get("optimizer"))

    @property
    def step_scheduler(self) -> StepSchedulerConfig:
        return _coerce(StepSchedulerConfig, self.raw.get("step_scheduler"))

    @property
    def checkpoint(self) -> CheckpointConfig:
        return _coerce(CheckpointConfig, self.raw.get("checkpoint"))

    @property
    def loss(self) -> LossConfig:
        return _coerce(LossConfig, self.raw.get("loss_fn"))
