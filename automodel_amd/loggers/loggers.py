"""Experiment logger configs: wandb / MLflow (optional deps, no-op offline).

Reference behavior: nemo_automodel/components/loggers/loggers.py:31-225
(WandbConfig/MLflowConfig/CometConfig built from YAML and instantiated on
rank 0). On this offline image the clients are not installed; configs degrade
to the JSONL MetricLogger which is always on.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

from automodel_amd.loggers.metric_logger import get_rank


@dataclass
class WandbConfig:
    project: str = "automodel_amd"
    name: str | None = None
    entity: str | None = None
    mode: str = "offline"

    def build(self, config: dict | None = None):
        if get_rank() != 0:
            return None
        try:
            import wandb
        except ImportError:
            return None
        return wandb.init(project=self.project, name=self.name, entity=self.entity,
                          mode=self.mode, config=config)


@dataclass
class MLflowConfig:
    tracking_uri: str | None = None
    experiment_name: str = "automodel_amd"
    run_name: str | None = None

    def build(self, config: dict | None = None):
        if get_rank() != 0:
            return None
        try:
            import mlflow
        except ImportError:
            return None
        if self.tracking_uri:
            mlflow.set_tracking_uri(self.tracking_uri)
        mlflow.set_experiment(self.experiment_name)
        run = mlflow.start_run(run_name=self.run_name)
        if config:
            mlflow.log_params({k: str(v)[:250] for k, v in config.items()})
        return run
