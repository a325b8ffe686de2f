import os

import pytest

from automodel_amd.config.loader import (
    ConfigNode,
    apply_overrides,
    load_yaml_config,
    parse_cli_overrides,
)


def test_attribute_and_item_access():
    cfg = ConfigNode({"a": {"b": 1}, "c": [1, {"d": 2}]})
    assert cfg.a.b == 1
    assert cfg["a"]["b"] == 1
    assert cfg.c[1].d == 2
    assert cfg.to_dict() == {"a": {"b": 1}, "c": [1, {"d": 2}]}


def test_env_interpolation(monkeypatch):
    monkeypatch.setenv("MY_TEST_VAR", "hello")
    cfg = ConfigNode({"x": "${MY_TEST_VAR}", "y": "${MISSING_VAR:fallback}"})
    assert cfg.x == "hello"
    assert cfg.y == "fallback"
    with pytest.raises(KeyError):
        ConfigNode({"z": "${DEFINITELY_MISSING_VAR}"})


def test_dotted_set_get():
    cfg = ConfigNode({})
    cfg.set_by_dotted("a.b.c", 42)
    assert cfg.get_by_dotted("a.b.c") == 42
    assert cfg.get_by_dotted("a.b.missing", "def") == "def"


def test_cli_overrides():
    ov = parse_cli_overrides(["--a.b=3", "--c", "true", "--d.e=[1,2]"])
    assert ov == {"a.b": 3, "c": True, "d.e": [1, 2]}
    cfg = ConfigNode({"a": {"b": 1}})
    apply_overrides(cfg, ov)
    assert cfg.a.b == 3 and cfg.c is True and cfg.d.e == [1, 2]


def test_target_instantiate():
    cfg = ConfigNode({
        "sched": {
            "_target_": "automodel_amd.training.step_scheduler.StepScheduler",
            "grad_acc_steps": 4,
            "max_steps": 10,
        }
    })
    obj = cfg.sched.instantiate()
    assert obj.grad_acc_steps == 4 and obj.max_steps == 10


def test_nested_instantiate_and_lazy():
    cfg = ConfigNode({
        "_target_": "builtins.dict",
        "inner": {"_target_": "builtins.list"},
        "lazy": {"_target_": "builtins.list", "_lazy_": True},
    })
    out = cfg.instantiate()
    assert out["inner"] == []
    assert isinstance(out["lazy"], ConfigNode)


def test_yaml_roundtrip(tmp_path):
    p = tmp_path / "c.yaml"
    p.write_text("model:\n  hidden: 64\nlist: [1, 2]\n")
    cfg = load_yaml_config(p)
    assert cfg.model.hidden == 64 and cfg.list == [1, 2]


def test_secret_redaction():
    cfg = ConfigNode({"wandb": {"api_key": "s3cret"}, "lr": 1.0})
    red = cfg.redacted_dict()
    assert red["wandb"]["api_key"] == "***" and red["lr"] == 1.0


def test_typed_recipe_config():
    from automodel_amd.recipes.typed_config import RecipeConfig

    cfg = ConfigNode({
        "distributed": {"tp": 2, "cp": 1},
        "optimizer": {"lr": 3e-4, "unknown_key_ignored": 1},
        "step_scheduler": {"max_steps": 7},
        "loss_fn": {"backend": "chunked", "chunk_size": 64},
    })
    rc = RecipeConfig(cfg)
    assert rc.distributed.tp == 2 and rc.distributed.dp_shard == -1
    assert rc.optimizer.lr == 3e-4
    sched = rc.step_scheduler.build()
    assert sched.max_steps == 7
    loss = rc.loss.build()
    assert loss.backend == "chunked"
