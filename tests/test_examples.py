"""Every example YAML must parse, resolve its recipe, and build its model
config on the meta device (catches config typos without running training)."""

import glob
import os

import pytest
import torch

from automodel_amd.cli.app import RECIPE_ALIASES
from automodel_amd.config.loader import load_yaml_config, resolve_target

EXAMPLES = sorted(glob.glob(os.path.join(os.path.dirname(__file__), "..",
                                         "examples", "**", "*.yaml"),
                            recursive=True))


@pytest.mark.parametrize("path", EXAMPLES, ids=[os.path.basename(p) for p in EXAMPLES])
def test_example_yaml_valid(path):
    cfg = load_yaml_config(path)
    recipe = cfg.get("recipe", "llm_finetune")
    target = RECIPE_ALIASES.get(recipe, recipe)
    cls = resolve_target(target)
    assert callable(cls)

    mcfg = cfg.get("model")
    assert mcfg is not None, "example must define a model"
    if mcfg.get("config") and "_lazy_" not in str(mcfg.to_dict()):
        from automodel_amd.models.registry import build_model

        model = build_model(
            config=mcfg.config.to_dict(),
            architecture=mcfg.get("architecture"),
            dtype="float32",
            meta_init=True,
        )
        assert sum(p.numel() for p in model.parameters()) > 0

    # distributed block must be internally consistent
    dist_cfg = cfg.get("distributed")
    if dist_cfg:
        for k in dist_cfg:
            assert k in ("dp_replicate", "dp_shard", "tp", "pp", "cp",
                         "sequence_parallel", "reshard_after_forward", "pipeline"), k


def test_examples_exist():
    assert len(EXAMPLES) >= 8
