""".env loading + typed env access (reference include/utils/env.hpp:13)."""

from __future__ import annotations

import os
from typing import Callable, Optional, TypeVar

T = TypeVar("T")


class EnvLoader:
    @staticmethod
    def load(path: str = ".env", override: bool = False):
        if not os.path.exists(path):
            return
        with open(path) as f:
            for line in f:
                line = line.strip()
                if not line or line.startswith("#") or "=" not in line:
                    continue
                k, _, v = line.partition("=")
                k, v = k.strip(), v.strip().strip('"').strip("'")
                if override or k not in os.environ:
                    os.environ[k] = v


def env_get(name: str, cast: Callable[[str], T] = str,
            default: Optional[T] = None) -> Optional[T]:
    v = os.environ.get(name)
    if v is None:
        return default
    if cast is bool:
        return v.lower() in ("1", "true", "yes", "on")  # type: ignore
    return cast(v)
