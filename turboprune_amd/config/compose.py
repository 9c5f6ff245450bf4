"""Hydra-surface-compatible YAML config composition, dependency-free.

The reference drives experiments through hydra (``@hydra.main(config_path=
"conf", config_name="config")`` + CLI overrides, reference:
run_experiment.py:21). hydra/omegaconf are not part of this framework's
runtime; this module reimplements the *surface* the reference exposes:

- a ``conf/`` tree of YAML files with group subdirectories
  (``dataset_params/``, ``optimizer_params/``, ...);
- top-level composite configs with a ``defaults:`` list
  (``- _self_``, ``- group: option`` entries);
- CLI overrides: ``a.b.c=value`` (set), ``+a.b=value`` (add),
  ``group=option`` (select a different group file),
  ``--config-name=NAME``.

Values are YAML-parsed, so ``lr=0.2``, ``distributed=false`` and
``milestones=[60,120]`` all work as with hydra.
"""

from __future__ import annotations

import copy
import os
from typing import Any, Iterable, Mapping, Optional

import yaml


class Config(dict):
    """A dict with attribute access and dotted-path get/set (omegaconf-lite)."""

    def __init__(self, data: Optional[Mapping[str, Any]] = None):
        super().__init__()
        if data:
            for k, v in data.items():
                self[k] = _wrap(v)

    # --- attribute access -------------------------------------------------
    def __getattr__(self, name: str) -> Any:
        try:
            return self[name]
        except KeyError as e:
            raise AttributeError(name) from e

    def __setattr__(self, name: str, value: Any) -> None:
        self[name] = _wrap(value)

    def __delattr__(self, name: str) -> None:
        del self[name]

    # --- dotted paths -----------------------------------------------------
    def select(self, path: str, default: Any = None) -> Any:
        node: Any = self
        for part in path.split("."):
            if isinstance(node, Mapping) and part in node:
                node = node[part]
            else:
                return default
        return node

    def set_path(self, path: str, value: Any, allow_new: bool = True) -> None:
        parts = path.split(".")
        node = self
        for part in parts[:-1]:
            if part not in node or not isinstance(node[part], Config):
                if not allow_new and part not in node:
                    raise KeyError(f"config path '{path}' does not exist "
                                   f"(missing '{part}'); use +{path}= to add")
                node[part] = Config()
            node = node[part]
        if not allow_new and parts[-1] not in node:
            raise KeyError(f"config key '{path}' does not exist; use +{path}= to add")
        node[parts[-1]] = _wrap(value)

    def merge(self, other: Mapping[str, Any]) -> None:
        for k, v in other.items():
            if k in self and isinstance(self[k], Config) and isinstance(v, Mapping):
                self[k].merge(v)
            else:
                self[k] = _wrap(v)

    def to_dict(self) -> dict:
        out: dict = {}
        for k, v in self.items():
            out[k] = v.to_dict() if isinstance(v, Config) else copy.deepcopy(v)
        return out

    def __deepcopy__(self, memo):
        return Config(self.to_dict())


_FLOAT_RE = None


def _coerce(v: Any) -> Any:
    """YAML 1.1 reads '5e-4' as a string; coerce such scalars to float
    (what YAML 1.2 / omegaconf would do)."""
    global _FLOAT_RE
    if isinstance(v, str):
        if _FLOAT_RE is None:
            import re
            _FLOAT_RE = re.compile(
                r"^[+-]?(\d+\.?\d*|\.\d+)[eE][+-]?\d+$")
        if _FLOAT_RE.match(v):
            return float(v)
    return v


def _wrap(v: Any) -> Any:
    if isinstance(v, Config):
        return v
    if isinstance(v, Mapping):
        return Config(v)
    if isinstance(v, list):
        return [_wrap(x) for x in v]
    return _coerce(v)


def load_yaml(path: str) -> Config:
    with open(path) as f:
        data = yaml.safe_load(f) or {}
    if not isinstance(data, dict):
        raise ValueError(f"{path} is not a mapping at top level")
    return Config(data)


def default_config_dir() -> str:
    """The in-repo conf/ tree (next to the package)."""
    return os.path.join(os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__)))), "conf")


def _parse_value(text: str) -> Any:
    try:
        return yaml.safe_load(text)
    except yaml.YAMLError:
        return text


def compose(config_name: str,
            overrides: Optional[Iterable[str]] = None,
            config_dir: Optional[str] = None) -> Config:
    """Compose a config like hydra would.

    ``config_name`` is the top-level composite (e.g. ``imagenet_er_balanced``)
    or ``config`` for the shipped default. Overrides are hydra-style strings.
    """
    config_dir = config_dir or default_config_dir()
    overrides = list(overrides or [])

    top_path = os.path.join(config_dir, f"{config_name}.yaml")
    if not os.path.exists(top_path):
        available = sorted(f[:-5] for f in os.listdir(config_dir)
                           if f.endswith(".yaml"))
        raise FileNotFoundError(
            f"config '{config_name}' not found in {config_dir}; "
            f"available: {available}")
    top = load_yaml(top_path)

    defaults = top.pop("defaults", None) or []
    # group selections from the defaults list
    group_choice: dict = {}
    order: list = []
    for entry in defaults:
        if entry == "_self_":
            order.append(("_self_", None))
        elif isinstance(entry, Mapping):
            for group, option in entry.items():
                group_choice[str(group)] = option
                order.append((str(group), option))
        else:
            order.append((str(entry), None))

    # group overrides (``pruning_params=iterative_imp``) change the selection
    kv_overrides = []
    for ov in overrides:
        if ov.startswith("--config-name"):
            continue
        add = ov.startswith("+")
        body = ov[1:] if add else ov
        if "=" not in body:
            raise ValueError(f"override '{ov}' has no '='")
        key, val = body.split("=", 1)
        if not add and "." not in key and key in group_choice:
            group_choice[key] = val.strip()
        else:
            kv_overrides.append((key, _parse_value(val), add))

    cfg = Config()
    for group, _ in order:
        if group == "_self_":
            cfg.merge(top)
            continue
        option = group_choice[group]
        if option in (None, "null"):
            continue
        sub_path = os.path.join(config_dir, group, f"{option}.yaml")
        if not os.path.exists(sub_path):
            raise FileNotFoundError(
                f"config group '{group}' has no option '{option}' "
                f"({sub_path} missing)")
        node = cfg
        for part in group.split("/"):
            if part not in node:
                node[part] = Config()
            node = node[part]
        node.merge(load_yaml(sub_path))
    if ("_self_", None) not in order:
        cfg.merge(top)

    for key, val, add in kv_overrides:
        # hydra semantics: plain key=value must name an existing key;
        # '+key=value' is the explicit add syntax
        cfg.set_path(key, val, allow_new=add)
    return cfg


def save_config(expt_dir: str, cfg: Config) -> None:
    """Dump the composed config to expt_config.yaml (reference:
    utils/harness_utils.py:148-156)."""
    with open(os.path.join(expt_dir, "expt_config.yaml"), "w") as f:
        yaml.dump(cfg.to_dict(), f, default_flow_style=False)


def parse_cli(argv: Iterable[str]) -> tuple:
    """Parse hydra-style CLI: ``--config-name=NAME`` (or ``-cn NAME``) plus
    override tokens. Returns (config_name, overrides)."""
    argv = list(argv)
    config_name = "config"
    overrides = []
    i = 0
    while i < len(argv):
        a = argv[i]
        if a.startswith("--config-name="):
            config_name = a.split("=", 1)[1]
        elif a in ("--config-name", "-cn"):
            i += 1
            config_name = argv[i]
        elif "=" in a:
            overrides.append(a)
        else:
            raise ValueError(f"unrecognized argument: {a}")
        i += 1
    return config_name, overrides
