"""Config system: flat dotted-key YAML with three-layer merge.

Reproduces the reference's config surface (ref train.py:30-56): a single
flat dict with dotted keys (``data.*``, ``lr.*``, ``model.*``, ``mpi.*``,
``loss.*``, ``training.*``), merged default -> dataset -> extra-JSON with
an assert-known-key rule, dumped to the workspace as ``params.yaml``.

Unlike the reference (which mutates the config dict with runtime state:
ref train.py:56,66,107-108,135-136), immutable configuration and runtime
state are separated: `Config` is the frozen merged mapping, `RuntimeState`
carries rank/logger/writer/epoch.
"""
from __future__ import annotations

import copy
import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, Iterator, Mapping, Optional

import yaml

_CONFIG_DIR = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "configs")

# Post-parse coercions (ref train.py:54-55): CSV strings -> int lists.
_CSV_INT_KEYS = ("training.gpus", "lr.decay_steps")


class Config(Mapping):
    """Immutable flat dotted-key config mapping."""

    def __init__(self, data: Dict[str, Any]):
        self._data = dict(data)

    # Mapping interface -------------------------------------------------
    def __getitem__(self, key: str) -> Any:
        return self._data[key]

    def __iter__(self) -> Iterator[str]:
        return iter(self._data)

    def __len__(self) -> int:
        return len(self._data)

    def get(self, key: str, default: Any = None) -> Any:
        return self._data.get(key, default)

    # ---------------------------------------------------------------
    def replace(self, **overrides: Any) -> "Config":
        """Return a new Config with dotted keys overridden (keys may be new)."""
        data = dict(self._data)
        for k, v in overrides.items():
            data[k] = v
        return Config(data)

    def updated(self, mapping: Dict[str, Any], allow_new: bool = False) -> "Config":
        data = dict(self._data)
        for k, v in mapping.items():
            if not allow_new and k not in data:
                raise KeyError(f"unknown config key: {k!r}")
            data[k] = v
        return Config(data)

    def to_dict(self) -> Dict[str, Any]:
        return copy.deepcopy(self._data)

    def dump_yaml(self, path: str) -> None:
        with open(path, "w") as f:
            yaml.safe_dump(self._serializable(), f, default_flow_style=False)

    def _serializable(self) -> Dict[str, Any]:
        out = {}
        for k, v in self._data.items():
            if isinstance(v, (list, tuple)):
                v = list(v)
            out[k] = v
        return out

    def __repr__(self) -> str:  # pragma: no cover
        return f"Config({self._data!r})"


def _coerce(config: Dict[str, Any]) -> Dict[str, Any]:
    for key in _CSV_INT_KEYS:
        if key in config and not isinstance(config[key], list):
            config[key] = [int(s) for s in str(config[key]).split(",")]
    return config


def merge_configs(default: Dict[str, Any],
                  dataset: Optional[Dict[str, Any]] = None,
                  extra: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
    """default -> dataset -> extra merge with unknown-key assertion
    (ref train.py:38-44)."""
    config = dict(default)
    for layer, name in ((dataset, "dataset config"), (extra, "extra_config")):
        if not layer:
            continue
        for k in layer:
            if k not in config:
                raise KeyError(f"unknown key {k!r} in {name}")
        config.update(layer)
    return config


def load_config(config_path: str,
                extra_config: str = "{}",
                default_path: Optional[str] = None) -> Config:
    """Load params_<dataset>.yaml over params_default.yaml over JSON extras."""
    if default_path is None:
        default_path = os.path.join(os.path.dirname(os.path.abspath(config_path)),
                                    "params_default.yaml")
        if not os.path.exists(default_path):
            default_path = os.path.join(_CONFIG_DIR, "params_default.yaml")
    with open(default_path, "r") as f:
        default = yaml.safe_load(f)
    dataset = None
    if config_path and os.path.abspath(config_path) != os.path.abspath(default_path):
        with open(config_path, "r") as f:
            dataset = yaml.safe_load(f)
    extra = json.loads(extra_config) if isinstance(extra_config, str) else dict(extra_config)
    merged = merge_configs(default, dataset, extra)
    return Config(_coerce(merged))


def default_config(**overrides: Any) -> Config:
    """The in-repo params_default.yaml, with overrides applied (new keys allowed)."""
    cfg = load_config(os.path.join(_CONFIG_DIR, "params_default.yaml"))
    return cfg.replace(**overrides) if overrides else cfg


@dataclass
class RuntimeState:
    """Mutable per-process runtime state (kept OUT of the Config)."""
    global_rank: int = 0
    local_rank: int = 0
    world_size: int = 1
    local_workspace: Optional[str] = None
    log_file: Optional[str] = None
    logger: Any = None
    tb_writer: Any = None
    current_epoch: int = 0
    global_step: int = 0
    extras: Dict[str, Any] = field(default_factory=dict)

    @property
    def is_rank0(self) -> bool:
        return self.global_rank == 0
