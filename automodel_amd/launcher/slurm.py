"""SLURM launcher: render + submit an sbatch script.

Reference behavior: the reference ships slurm.sub at repo root (container +
srun torchrun) and a NeMo-Run SLURM executor (components/launcher/nemo_run/).
This renders an equivalent sbatch driving torch.distributed.run with RCCL
env over xGMI; submission shells out to sbatch when present.
"""

from __future__ import annotations

import os
import subprocess

SBATCH_TEMPLATE = """#!/bin/bash
#SBATCH --job-name={job_name}
#SBATCH --nodes={nodes}
#SBATCH --ntasks-per-node=1
#SBATCH --gpus-per-node={gpus_per_node}
#SBATCH --time={time_limit}
{extra_directives}
export MASTER_ADDR=$(scontrol show hostnames $SLURM_JOB_NODELIST | head -n1)
export MASTER_PORT={master_port}
export HSA_ENABLE_IPC_MODE_LEGACY=0
srun --kill-on-bad-exit=1 python -m torch.distributed.run \\
  --nnodes={nodes} --nproc-per-node={gpus_per_node} \\
  --rdzv-backend=c10d --rdzv-endpoint=$MASTER_ADDR:$MASTER_PORT \\
  -m automodel_amd.launcher.interactive {cfg_path} {recipe_target} {overrides}
"""


class SlurmLauncher:
    def __init__(self, nodes: int = 1, gpus_per_node: int = 8,
                 job_name: str = "automodel_amd", time_limit: str = "04:00:00",
                 master_port: int = 29512, account: str | None = None,
                 partition: str | None = None):
        self.nodes = nodes
        self.gpus_per_node = gpus_per_node
        self.job_name = job_name
        self.time_limit = time_limit
        self.master_port = master_port
        self.account = account
        self.partition = partition

    def render(self, cfg_path: str, recipe_target: str, overrides: list[str]) -> str:
        extra = []
        if self.account:
            extra.append(f"#SBATCH --account={self.account}")
        if self.partition:
            extra.append(f"#SBATCH --partition={self.partition}")
        return SBATCH_TEMPLATE.format(
            job_name=self.job_name, nodes=self.nodes,
            gpus_per_node=self.gpus_per_node, time_limit=self.time_limit,
            master_port=self.master_port, extra_directives="\n".join(extra),
            cfg_path=cfg_path, recipe_target=recipe_target,
            overrides=" ".join(overrides),
        )

    def launch(self, cfg_path: str, recipe_target: str, overrides: list[str],
               script_path: str = "automodel_job.sub", submit: bool = True) -> str:
        script = self.render(cfg_path, recipe_target, overrides)
        with open(script_path, "w") as f:
            f.write(script)
        if submit and _which("sbatch"):
            subprocess.run(["sbatch", script_path], check=True)
        return script_path


def _which(prog: str) -> str | None:
    for d in os.environ.get("PATH", "").split(os.pathsep):
        p = os.path.join(d, prog)
        if os.path.isfile(p) and os.access(p, os.X_OK):
            return p
    return None
