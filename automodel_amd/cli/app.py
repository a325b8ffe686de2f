"""``automodel`` CLI: YAML in, recipe out.

Reference behavior: nemo_automodel/cli/app.py:95-159 (parse YAML, resolve the
``recipe:`` target, dispatch to a launcher). Usage:

    automodel <cfg.yaml> [--nproc-per-node N] [--a.b.c=v ...]
"""

from __future__ import annotations

import sys

from automodel_amd.config.loader import (
    apply_overrides,
    load_yaml_config,
    parse_cli_overrides,
    resolve_target,
)

RECIPE_ALIASES = {
    "llm_finetune": "automodel_amd.recipes.llm.train_ft.TrainFinetuneRecipeForNextTokenPrediction",
    "llm_pretrain": "automodel_amd.recipes.llm.train_ft.TrainFinetuneRecipeForNextTokenPrediction",
    "llm_benchmark": "automodel_amd.recipes.llm.benchmark.BenchmarkingRecipeForNextTokenPrediction",
    "vlm_finetune": "automodel_amd.recipes.vlm.finetune.FinetuneRecipeForVLM",
    "llm_kd": "automodel_amd.recipes.llm.kd.KDRecipeForNextTokenPrediction",
    "llm_seq_cls": "automodel_amd.recipes.llm.train_seq_cls.TrainFinetuneRecipeForSequenceClassification",
    "retrieval": "automodel_amd.recipes.llm.train_retrieval.TrainRecipeForRetrieval",
    "eagle_draft": "automodel_amd.speculative.train_draft.TrainEagleDraftRecipe",
    "llm_dllm": "automodel_amd.recipes.llm.train_dllm.TrainDiffusionLMRecipe",
}


def query_capabilities(arch: str | None = None) -> None:
    """Print per-model capability flags (reference cli/query_capabilities.py)."""
    from automodel_amd.models.common.capabilities import _CAPS, ModelCapabilities
    from automodel_amd.models.registry import _REGISTRY, _ensure_builtin

    _ensure_builtin()
    rows = [arch] if arch else sorted(set(list(_REGISTRY) + list(_CAPS)))
    for name in rows:
        cls = _REGISTRY.get(name)
        caps = _CAPS.get(cls.__name__ if cls else name, ModelCapabilities())
        flags = ", ".join(f"{k}={v}" for k, v in vars(caps).items())
        print(f"{name}: {flags}")


def main(argv: list[str] | None = None) -> None:
    argv = argv if argv is not None else sys.argv[1:]
    if not argv or argv[0] in ("-h", "--help"):
        print(__doc__)
        return
    if argv[0] == "capabilities":
        query_capabilities(argv[1] if len(argv) > 1 else None)
        return
    cfg_path = argv[0]
    rest = argv[1:]
    nproc = 1
    if "--nproc-per-node" in rest:
        i = rest.index("--nproc-per-node")
        nproc = int(rest[i + 1])
        rest = rest[:i] + rest[i + 2:]
    overrides = parse_cli_overrides(rest)

    cfg = load_yaml_config(cfg_path)
    apply_overrides(cfg, overrides)
    recipe_name = cfg.get("recipe", "llm_finetune")
    target = RECIPE_ALIASES.get(recipe_name, recipe_name)

    from automodel_amd.launcher.interactive import InteractiveLauncher

    InteractiveLauncher(nproc_per_node=nproc).launch(cfg_path, target, rest)


if __name__ == "__main__":
    main()
