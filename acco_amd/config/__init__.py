"""Hydra-style config system (no external deps).

Mirrors the reference's Hydra surface (reference config/config.yaml:1-12 and
main.py:25): config groups ``data``, ``train``, ``model`` with defaults
``openwebtext``/``acco``/``gptneo``, selected on the command line as
``train=acco data=openwebtext model=gptneo`` plus dotted overrides such as
``train.nb_steps_tot=100``.

Implemented natively because the target environment has no Hydra/OmegaConf;
the CLI syntax and the flag surface (reference config/train/acco.yaml:1-28)
are kept identical.
"""

from __future__ import annotations

import copy
import json
import os
from typing import Any, Dict, List

import yaml

CONFIG_DIR = os.path.dirname(os.path.abspath(__file__))

GROUPS = ("data", "train", "model")


class Cfg(dict):
    """A dict with attribute access, nested. Similar surface to OmegaConf's
    DictConfig for the subset the trainer uses (attr access + to_container)."""

    def __getattr__(self, k: str) -> Any:
        try:
            return self[k]
        except KeyError as e:
            raise AttributeError(k) from e

    def __setattr__(self, k: str, v: Any) -> None:
        self[k] = v

    @staticmethod
    def wrap(obj: Any) -> Any:
        if isinstance(obj, dict):
            return Cfg({k: Cfg.wrap(v) for k, v in obj.items()})
        if isinstance(obj, list):
            return [Cfg.wrap(v) for v in obj]
        return obj

    def to_container(self) -> Dict[str, Any]:
        def unwrap(o: Any) -> Any:
            if isinstance(o, dict):
                return {k: unwrap(v) for k, v in o.items()}
            if isinstance(o, list):
                return [unwrap(v) for v in o]
            return o

        return unwrap(self)


def _parse_value(s: str) -> Any:
    """Parse a CLI override value with YAML scalar semantics."""
    try:
        return yaml.safe_load(s)
    except yaml.YAMLError:
        return s


def _load_group_file(group: str, name: str) -> Dict[str, Any]:
    path = os.path.join(CONFIG_DIR, group, name + ".yaml")
    if not os.path.exists(path):
        path_json = os.path.join(CONFIG_DIR, group, name + ".json")
        if os.path.exists(path_json):
            with open(path_json) as f:
                return json.load(f)
        raise FileNotFoundError(
            f"config group '{group}' has no preset '{name}' "
            f"(looked in {os.path.join(CONFIG_DIR, group)})"
        )
    with open(path) as f:
        return yaml.safe_load(f) or {}


def _set_dotted(cfg: Dict[str, Any], key: str, value: Any) -> None:
    parts = key.split(".")
    node = cfg
    for p in parts[:-1]:
        if p not in node or not isinstance(node[p], dict):
            node[p] = {}
        node = node[p]
    node[parts[-1]] = value


def load_config(overrides: List[str] | None = None) -> Cfg:
    """Load the root config, apply group selections and dotted overrides.

    ``overrides`` is e.g. ``["train=acco", "model=llama-1b",
    "train.nb_steps_tot=100"]``.
    """
    overrides = list(overrides or [])
    with open(os.path.join(CONFIG_DIR, "config.yaml")) as f:
        root = yaml.safe_load(f)

    defaults = root.pop("defaults", {})
    group_sel = {g: defaults.get(g) for g in GROUPS}

    # First pass: group selections (bare `group=name`).
    dotted: List[tuple] = []
    for ov in overrides:
        if "=" not in ov:
            raise ValueError(f"override '{ov}' is not of the form key=value")
        key, _, val = ov.partition("=")
        if key in GROUPS and "." not in key:
            group_sel[key] = val
        else:
            dotted.append((key, _parse_value(val)))

    cfg: Dict[str, Any] = copy.deepcopy(root)
    for g in GROUPS:
        name = group_sel[g]
        if name is None:
            continue
        cfg[g] = _load_group_file(g, str(name))
        cfg[g]["_name"] = str(name)

    for key, val in dotted:
        _set_dotted(cfg, key, val)

    return Cfg.wrap(cfg)
