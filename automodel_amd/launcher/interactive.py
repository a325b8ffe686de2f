"""Interactive launcher: run in-process or exec torchrun.

Reference behavior: nemo_automodel/components/launcher/interactive.py:70-141
(if already a torchrun worker, run the recipe in-process; else exec
``torchrun --nproc-per-node N`` over this module).
"""

from __future__ import annotations

import os
import subprocess
import sys

from automodel_amd.config.loader import (
    apply_overrides,
    load_yaml_config,
    parse_cli_overrides,
    resolve_target,
)


def _is_torchrun_worker() -> bool:
    return "RANK" in os.environ and "WORLD_SIZE" in os.environ


class InteractiveLauncher:
    def __init__(self, nproc_per_node: int = 1):
        self.nproc_per_node = nproc_per_node

    def launch(self, cfg_path: str, recipe_target: str, overrides: list[str]) -> None:
        if self.nproc_per_node <= 1 or _is_torchrun_worker():
            self._run_in_process(cfg_path, recipe_target, overrides)
            return
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", f"--nproc-per-node={self.nproc_per_node}",
            "--master-addr", "127.0.0.1",
            "-m", "automodel_amd.launcher.interactive",
            cfg_path, recipe_target, *overrides,
        ]
        raise SystemExit(subprocess.call(cmd))

    @staticmethod
    def _run_in_process(cfg_path: str, recipe_target: str, overrides: list[str]) -> None:
        cfg = load_yaml_config(cfg_path)
        apply_overrides(cfg, parse_cli_overrides(overrides))
        recipe_cls = resolve_target(recipe_target)
        recipe = recipe_cls(cfg)
        recipe.setup()
        recipe.run_train_validation_loop()


if __name__ == "__main__":
    # torchrun re-entry point: argv = [cfg_path, recipe_target, overrides...]
    InteractiveLauncher()._run_in_process(sys.argv[1], sys.argv[2], sys.argv[3:])
