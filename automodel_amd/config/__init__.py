from automodel_amd.config.loader import (
    ConfigNode,
    apply_overrides,
    load_yaml_config,
    parse_cli_overrides,
    resolve_target,
)

__all__ = [
    "ConfigNode",
    "apply_overrides",
    "load_yaml_config",
    "parse_cli_overrides",
    "resolve_target",
]
