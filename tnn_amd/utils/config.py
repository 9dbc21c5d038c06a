"""Typed JSON config store (reference include/common/config.hpp:16-90).

The reference's ``TConfig`` preserves exact C++ value types for lossless
round-trips; in Python JSON types are already faithful, so this is a thin
dict wrapper with dotted-path access, defaults and JSON round-trip.
"""

from __future__ import annotations

import json
from typing import Any, Dict, Optional


class TConfig:
    def __init__(self, data: Optional[Dict[str, Any]] = None):
        self._data: Dict[str, Any] = dict(data or {})

    def set(self, key: str, value: Any) -> "TConfig":
        parts = key.split(".")
        d = self._data
        for p in parts[:-1]:
            d = d.setdefault(p, {})
        d[parts[-1]] = value
        return self

    def get(self, key: str, default: Any = None) -> Any:
        d: Any = self._data
        for p in key.split("."):
            if not isinstance(d, dict) or p not in d:
                return default
            d = d[p]
        return d

    def has(self, key: str) -> bool:
        sentinel = object()
        return self.get(key, sentinel) is not sentinel

    def to_dict(self) -> Dict[str, Any]:
        return self._data

    def to_json(self, **kw) -> str:
        return json.dumps(self._data, **kw)

    @classmethod
    def from_json(cls, text: str) -> "TConfig":
        return cls(json.loads(text))

    @classmethod
    def from_file(cls, path: str) -> "TConfig":
        with open(path) as f:
            return cls(json.load(f))

    def save(self, path: str):
        with open(path, "w") as f:
            json.dump(self._data, f, indent=2)

    def __eq__(self, other):
        return isinstance(other, TConfig) and self._data == other._data

    def __repr__(self):
        return f"TConfig({self._data!r})"
