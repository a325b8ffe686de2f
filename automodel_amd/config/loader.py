"""YAML configuration system.

MI355X-native re-design of the reference's config layer
(reference: nemo_automodel/components/config/loader.py:336-810 — ConfigNode with
Hydra-style ``_target_`` instantiation, ``${ENV}`` interpolation and dotted
overrides). The behavior (YAML -> attribute-accessible tree -> recursive
``instantiate``) is kept; the implementation is new and minimal.
"""

from __future__ import annotations

import importlib
import os
import re
from typing import Any, Iterator, Mapping

import yaml

__all__ = ["ConfigNode", "load_yaml_config", "parse_cli_overrides", "translate_value"]

_ENV_RE = re.compile(r"\$\{([A-Za-z_][A-Za-z0-9_]*)(?::([^}]*))?\}")

# Keys whose values are redacted when dumping a config (reference: loader.py:109).
_SECRET_RE = re.compile(r"(api_key|token|secret|password)", re.IGNORECASE)


def _resolve_env(value: str) -> str:
    """Interpolate ``${VAR}`` / ``${VAR:default}`` using os.environ."""

    def repl(m: re.Match) -> str:
        var, default = m.group(1), m.group(2)
        if var in os.environ:
            return os.environ[var]
        if default is not None:
            return default
        raise KeyError(f"environment variable '{var}' referenced in config is not set")

    return _ENV_RE.sub(repl, value)


def resolve_target(path: str) -> Any:
    """Import a dotted path like ``automodel_amd.optim.build_adamw``.

    Tries progressively shorter module prefixes so class attributes and
    nested names resolve (``pkg.mod.Class.method``).
    """
    parts = path.split(".")
    for split in range(len(parts) - 1, 0, -1):
        module_name = ".".join(parts[:split])
        try:
            obj = importlib.import_module(module_name)
        except ImportError:
            continue
        try:
            for attr in parts[split:]:
                obj = getattr(obj, attr)
        except AttributeError:
            continue
        return obj
    # Last resort: maybe it's a plain module.
    try:
        return importlib.import_module(path)
    except ImportError as e:
        raise ImportError(f"cannot resolve _target_ '{path}'") from e


def translate_value(text: str) -> Any:
    """Parse a CLI override value string into a python object (YAML rules)."""
    try:
        return yaml.safe_load(text)
    except yaml.YAMLError:
        return text


class ConfigNode:
    """An attribute-accessible config tree backed by a dict.

    Supports: attribute and item access, ``get``, ``to_dict``, dotted
    ``set_by_dotted``/``get_by_dotted``, and recursive ``instantiate`` of
    ``_target_`` nodes.
    """

    def __init__(self, data: Mapping[str, Any] | None = None):
        object.__setattr__(self, "_data", {})
        if data:
            for k, v in data.items():
                self._data[k] = self._wrap(v)

    # -- construction helpers -------------------------------------------------
    @classmethod
    def _wrap(cls, v: Any) -> Any:
        if isinstance(v, ConfigNode):
            return v
        if isinstance(v, Mapping):
            return cls(v)
        if isinstance(v, list):
            return [cls._wrap(x) for x in v]
        if isinstance(v, str):
            return _resolve_env(v)
        return v

    @staticmethod
    def _unwrap(v: Any) -> Any:
        if isinstance(v, ConfigNode):
            return v.to_dict()
        if isinstance(v, list):
            return [ConfigNode._unwrap(x) for x in v]
        return v

    # -- mapping protocol ------------------------------------------------------
    def __getattr__(self, name: str) -> Any:
        data = object.__getattribute__(self, "_data")
        if name in data:
            return data[name]
        raise AttributeError(f"config has no key '{name}'")

    def __setattr__(self, name: str, value: Any) -> None:
        self._data[name] = self._wrap(value)

    def __getitem__(self, key: str) -> Any:
        return self._data[key]

    def __setitem__(self, key: str, value: Any) -> None:
        self._data[key] = self._wrap(value)

    def __contains__(self, key: str) -> bool:
        return key in self._data

    def __iter__(self) -> Iterator[str]:
        return iter(self._data)

    def __len__(self) -> int:
        return len(self._data)

    def __eq__(self, other: object) -> bool:
        if isinstance(other, ConfigNode):
            return self.to_dict() == other.to_dict()
        if isinstance(other, Mapping):
            return self.to_dict() == dict(other)
        return NotImplemented

    def __repr__(self) -> str:
        return f"ConfigNode({self.to_dict()!r})"

    def keys(self):
        return self._data.keys()

    def items(self):
        return self._data.items()

    def values(self):
        return self._data.values()

    def get(self, key: str, default: Any = None) -> Any:
        return self._data.get(key, default)

    def pop(self, key: str, *default: Any) -> Any:
        return self._data.pop(key, *default)

    def setdefault(self, key: str, default: Any = None) -> Any:
        if key not in self._data:
            self._data[key] = self._wrap(default)
        return self._data[key]

    def to_dict(self) -> dict:
        return {k: self._unwrap(v) for k, v in self._data.items()}

    def redacted_dict(self) -> dict:
        """to_dict with secret-looking values replaced (reference loader.py:109)."""

        def scrub(d: Any) -> Any:
            if isinstance(d, dict):
                return {
                    k: ("***" if _SECRET_RE.search(k) and isinstance(v, str) else scrub(v))
                    for k, v in d.items()
                }
            if isinstance(d, list):
                return [scrub(x) for x in d]
            return d

        return scrub(self.to_dict())

    # -- dotted access ---------------------------------------------------------
    def get_by_dotted(self, dotted: str, default: Any = None) -> Any:
        node: Any = self
        for part in dotted.split("."):
            if isinstance(node, ConfigNode) and part in node:
                node = node[part]
            else:
                return default
        return node

    def set_by_dotted(self, dotted: str, value: Any) -> None:
        parts = dotted.split(".")
        node = self
        for part in parts[:-1]:
            if part not in node or not isinstance(node[part], ConfigNode):
                node[part] = ConfigNode()
            node = node[part]
        node[parts[-1]] = value

    # -- instantiation ---------------------------------------------------------
    def instantiate(self, **overrides: Any) -> Any:
        """Build the object described by this node's ``_target_``.

        Child nodes that themselves carry ``_target_`` are instantiated
        recursively unless marked ``_lazy_: true`` (then the node is passed
        through as-is for the callee to build).
        """
        if "_target_" not in self._data:
            raise ValueError("instantiate() called on a node without _target_")
        target = resolve_target(self._data["_target_"])
        kwargs: dict[str, Any] = {}
        for k, v in self._data.items():
            if k in ("_target_", "_lazy_"):
                continue
            kwargs[k] = _maybe_instantiate(v)
        kwargs.update(overrides)
        return target(**kwargs)

    def maybe_instantiate(self, default: Any = None, **overrides: Any) -> Any:
        if "_target_" in self._data:
            return self.instantiate(**overrides)
        return default


def _maybe_instantiate(v: Any) -> Any:
    if isinstance(v, ConfigNode):
        if "_target_" in v and not v.get("_lazy_", False):
            return v.instantiate()
        return v
    if isinstance(v, list):
        return [_maybe_instantiate(x) for x in v]
    return v


def load_yaml_config(path: str | os.PathLike) -> ConfigNode:
    with open(path) as f:
        data = yaml.safe_load(f) or {}
    if not isinstance(data, dict):
        raise ValueError(f"top level of {path} must be a mapping")
    return ConfigNode(data)


def parse_cli_overrides(argv: list[str]) -> dict[str, Any]:
    """Parse ``--a.b.c=v`` / ``--a.b.c v`` pairs into a dotted->value dict.

    Reference behavior: nemo_automodel/components/config/_arg_parser.py:79.
    """
    out: dict[str, Any] = {}
    i = 0
    while i < len(argv):
        tok = argv[i]
        if not tok.startswith("--"):
            raise ValueError(f"unexpected argument '{tok}' (expected --key=value)")
        key = tok[2:]
        if "=" in key:
            key, _, val = key.partition("=")
        else:
            if i + 1 >= len(argv):
                raise ValueError(f"missing value for override '{tok}'")
            i += 1
            val = argv[i]
        out[key] = translate_value(val)
        i += 1
    return out


def apply_overrides(cfg: ConfigNode, overrides: Mapping[str, Any]) -> ConfigNode:
    for k, v in overrides.items():
        cfg.set_by_dotted(k, v)
    return cfg
