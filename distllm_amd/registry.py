"""Per-process warm-start cache for expensive objects (models).

Capability parity with reference distllm/registry.py:44-207: a size-1
cache keyed on constructor arguments that keeps ONE live model per
process across task invocations, with shutdown callbacks. In the MI355X
framework this backs long-lived per-rank model residency (the reference
used it per Parsl worker process).
"""

from __future__ import annotations

import hashlib
import json
from typing import Any, Callable


def _freeze(obj: Any) -> str:
    try:
        return json.dumps(obj, sort_keys=True, default=repr)
    except TypeError:
        return repr(obj)


def _key(fn: Callable, args: tuple, kwargs: dict) -> str:
    payload = f"{fn.__module__}.{fn.__qualname__}|{_freeze(args)}|{_freeze(kwargs)}"
    return hashlib.sha256(payload.encode()).hexdigest()


class RegistrySingleton:
    """Holds one live object per process; replaces it when the key changes."""

    _instance: "RegistrySingleton | None" = None

    def __new__(cls) -> "RegistrySingleton":
        if cls._instance is None:
            cls._instance = super().__new__(cls)
            cls._instance._cache_key = None
            cls._instance._cache_obj = None
            cls._instance._shutdown = None
        return cls._instance

    def get(
        self,
        factory: Callable[..., Any],
        *args: Any,
        shutdown_callback: Callable[[Any], None] | None = None,
        **kwargs: Any,
    ) -> Any:
        key = _key(factory, args, kwargs)
        if key != self._cache_key:
            self.clear()
            self._cache_obj = factory(*args, **kwargs)
            self._cache_key = key
            self._shutdown = shutdown_callback
        return self._cache_obj

    def clear(self) -> None:
        if self._cache_obj is not None and self._shutdown is not None:
            try:
                self._shutdown(self._cache_obj)
            except Exception:
                pass
        self._cache_key = None
        self._cache_obj = None
        self._shutdown = None


registry = RegistrySingleton()


def register(shutdown_callback: Callable[[Any], None] | None = None):
    """Decorator form: memoize a factory through the registry."""

    def deco(fn: Callable[..., Any]) -> Callable[..., Any]:
        import functools

        @functools.wraps(fn)
        def wrapped(*args: Any, **kwargs: Any) -> Any:
            return registry.get(
                fn, *args, shutdown_callback=shutdown_callback, **kwargs
            )

        return wrapped

    return deco
