"""Options: YAML-backed key/value store.

Mirrors the reference's Options class (ref: src/common/options.{h,cpp} and
src/common/fastopt.{h,cpp}): every CLI flag and config-file key lives in one
flat dict keyed by the long option name (dashes, not underscores), with typed
access helpers.
"""
from __future__ import annotations

import copy
from typing import Any, Dict, Iterable

import yaml


class Options:
    """Flat key->value store with typed access, as the reference uses everywhere."""

    def __init__(self, values: Dict[str, Any] | None = None):
        self._values: Dict[str, Any] = dict(values or {})

    # -- mutation ---------------------------------------------------------
    def set(self, key: str, value: Any) -> None:
        self._values[key] = value

    def update(self, other: Dict[str, Any]) -> None:
        self._values.update(other)

    # -- access -----------------------------------------------------------
    def has(self, key: str) -> bool:
        return key in self._values

    def get(self, key: str, default: Any = None) -> Any:
        v = self._values.get(key, default)
        return v

    def __getitem__(self, key: str) -> Any:
        if key not in self._values:
            raise KeyError(f"Option '{key}' has not been defined")
        return self._values[key]

    def __contains__(self, key: str) -> bool:
        return key in self._values

    def bool(self, key: str) -> bool:
        return bool(self[key])

    def int(self, key: str) -> int:
        return int(self[key])

    def float(self, key: str) -> float:
        return float(self[key])

    def str(self, key: str) -> str:
        v = self.get(key)
        return "" if v is None else str(v)

    def list(self, key: str) -> list:
        v = self.get(key)
        if v is None:
            return []
        if isinstance(v, (list, tuple)):
            return list(v)
        return [v]

    def keys(self) -> Iterable[str]:
        return self._values.keys()

    def as_dict(self) -> Dict[str, Any]:
        return copy.deepcopy(self._values)

    def clone(self) -> "Options":
        return Options(self.as_dict())

    # -- serialization ----------------------------------------------------
    def to_yaml(self) -> str:
        return yaml.safe_dump(self._values, default_flow_style=False, sort_keys=True)

    @classmethod
    def from_yaml(cls, text: str) -> "Options":
        data = yaml.safe_load(text) or {}
        return cls(data)

    def __repr__(self) -> str:  # pragma: no cover
        return f"Options({len(self._values)} keys)"
