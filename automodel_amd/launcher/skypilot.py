"""SkyPilot launcher: render a task YAML and submit via the `sky` CLI.

Reference behavior: nemo_automodel/components/launcher/skypilot/
{config.py,launcher.py,utils.py} (SkyPilotConfig dataclass, torchrun command
built from SKYPILOT_NUM_NODES/NODE_RANK/NODE_IPS, workdir sync, sky.launch).
Here the task is rendered as the declarative SkyPilot task YAML (portable:
works with `sky launch task.yaml` on any machine with sky installed — this
training container has no cloud access, so rendering is the tested product
and submission shells out only when `sky` exists). MI355X specifics baked
into the task: ROCm accelerator name, HSA_ENABLE_IPC_MODE_LEGACY=0 for
dmabuf IPC, and RCCL over xGMI needs no IB env inside one node.
"""

from __future__ import annotations

import os
import subprocess
from dataclasses import dataclass, field

import yaml

SUPPORTED_CLOUDS = ("aws", "gcp", "azure", "lambda", "kubernetes", "oci")


@dataclass
class SkyPilotConfig:
    cloud: str = "kubernetes"
    accelerators: str = "MI355X:8"
    num_nodes: int = 1
    use_spot: bool = False
    disk_size: int = 256
    instance_type: str | None = None
    region: str | None = None
    zone: str | None = None
    job_name: str = "automodel-amd"
    setup: str = "cd ~/sky_workdir && pip install -e . --quiet && python -m automodel_amd.ops.build"
    env_vars: dict = field(default_factory=dict)
    master_port: int = 29512

    def __post_init__(self) -> None:
        if self.cloud.lower() not in SUPPORTED_CLOUDS:
            raise ValueError(f"cloud must be one of {SUPPORTED_CLOUDS}, got {self.cloud!r}")
        if self.num_nodes < 1:
            raise ValueError(f"num_nodes must be >= 1, got {self.num_nodes}")

    @property
    def gpus_per_node(self) -> int:
        if ":" in self.accelerators:
            return int(self.accelerators.rsplit(":", 1)[1])
        return 1


class SkyPilotLauncher:
    def __init__(self, **kwargs):
        self.config = SkyPilotConfig(**kwargs)

    def build_command(self, cfg_path: str, recipe_target: str,
                      overrides: list[str] | None = None) -> str:
        c = self.config
        parts = [
            "cd ~/sky_workdir &&",
            f"python -m torch.distributed.run --nproc-per-node={c.gpus_per_node}",
        ]
        if c.num_nodes > 1:
            parts += [
                "--nnodes=$SKYPILOT_NUM_NODES",
                "--node-rank=$SKYPILOT_NODE_RANK",
                "--rdzv-backend=c10d",
                f"--rdzv-endpoint=$(echo \"$SKYPILOT_NODE_IPS\" | head -n1):{c.master_port}",
            ]
        else:
            parts += ["--standalone"]
        parts += ["-m", "automodel_amd.launcher.interactive", cfg_path, recipe_target]
        parts += overrides or []
        return " ".join(parts)

    def render_task(self, cfg_path: str, recipe_target: str,
                    overrides: list[str] | None = None) -> dict:
        """The SkyPilot task spec as a dict (dump with yaml for `sky launch`)."""
        c = self.config
        resources: dict = {
            "cloud": c.cloud,
            "accelerators": c.accelerators,
            "use_spot": c.use_spot,
            "disk_size": c.disk_size,
        }
        for k in ("instance_type", "region", "zone"):
            v = getattr(c, k)
            if v:
                resources[k] = v
        envs = {"HSA_ENABLE_IPC_MODE_LEGACY": "0", **c.env_vars}
        return {
            "name": c.job_name,
            "num_nodes": c.num_nodes,
            "workdir": ".",
            "resources": resources,
            "envs": envs,
            "setup": c.setup,
            "run": self.build_command(cfg_path, recipe_target, overrides),
        }

    def launch(self, cfg_path: str, recipe_target: str,
               overrides: list[str] | None = None,
               task_path: str = "skypilot_task.yaml", submit: bool = True) -> str:
        task = self.render_task(cfg_path, recipe_target, overrides)
        with open(task_path, "w") as f:
            yaml.safe_dump(task, f, sort_keys=False)
        if submit and _which("sky"):
            subprocess.run(["sky", "launch", "-y", task_path], check=True)
        return task_path


def _which(prog: str) -> str | None:
    for d in os.environ.get("PATH", "").split(os.pathsep):
        p = os.path.join(d, prog)
        if os.path.isfile(p) and os.access(p, os.X_OK):
            return p
    return None
