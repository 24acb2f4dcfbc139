"""Foundation utilities: config base, batching, timing.

Capability parity with reference distllm/utils.py:20-128 and
distllm/timer.py:36-162 (BaseConfig YAML/JSON round-trip, batch_data,
Timer/TimeLogger with parseable `[timer]` lines) — re-implemented, not
copied; the `[timer]` stdout line format is kept log-compatible.
"""

from __future__ import annotations

import json
import re
import time
from pathlib import Path
from typing import Any, Iterator, TypeVar

import yaml
from pydantic import BaseModel, ConfigDict

T = TypeVar("T")

PathLike = str | Path


class BaseConfig(BaseModel):
    """Pydantic base for every config in the framework.

    Subclasses carry a ``name: Literal[...]`` discriminator so YAML
    ``name:`` fields select the implementation (same user-facing schema
    as the reference's strategy configs).
    """

    model_config = ConfigDict(extra="forbid")

    @classmethod
    def from_yaml(cls, path: PathLike) -> "BaseConfig":
        with open(path) as fh:
            data = yaml.safe_load(fh)
        return cls(**(data or {}))

    def write_yaml(self, path: PathLike) -> None:
        Path(path).parent.mkdir(parents=True, exist_ok=True)
        with open(path, "w") as fh:
            yaml.safe_dump(
                json.loads(self.model_dump_json()), fh, sort_keys=False
            )

    @classmethod
    def from_json(cls, path: PathLike) -> "BaseConfig":
        with open(path) as fh:
            return cls(**json.load(fh))

    def write_json(self, path: PathLike) -> None:
        Path(path).parent.mkdir(parents=True, exist_ok=True)
        with open(path, "w") as fh:
            fh.write(self.model_dump_json(indent=2))


def batch_data(data: list[T], batch_size: int) -> list[list[T]]:
    """Split ``data`` into consecutive chunks of at most ``batch_size``."""
    if batch_size < 1:
        raise ValueError(f"batch_size must be >= 1, got {batch_size}")
    return [data[i : i + batch_size] for i in range(0, len(data), batch_size)]


def iter_batches(data: list[T], batch_size: int) -> Iterator[list[T]]:
    for i in range(0, len(data), batch_size):
        yield data[i : i + batch_size]


# ---------------------------------------------------------------------------
# Timing
# ---------------------------------------------------------------------------

_TIMER_RE = re.compile(
    r"\[timer\] \[(?P<tags>.*?)\] in (?P<elapsed>[0-9.eE+-]+) s "
    r"\(start: (?P<start>[0-9.]+), end: (?P<end>[0-9.]+)\)"
)


class Timer:
    """Context-manager wall timer emitting parseable ``[timer]`` lines.

    Log-line format is compatible with the reference's TimeLogger
    (distllm/timer.py:126-162) so downstream log parsers keep working.
    """

    def __init__(self, *tags: str, logfile: PathLike | None = None) -> None:
        self.tags = [str(t) for t in tags]
        self.logfile = logfile
        self.start_unix = 0.0
        self.end_unix = 0.0
        self._t0 = 0
        self.elapsed_s = 0.0

    def __enter__(self) -> "Timer":
        self.start_unix = time.time()
        self._t0 = time.perf_counter_ns()
        return self

    def __exit__(self, *exc: Any) -> None:
        self.elapsed_s = (time.perf_counter_ns() - self._t0) / 1e9
        self.end_unix = time.time()
        line = (
            f"[timer] [{', '.join(self.tags)}] in {self.elapsed_s} s "
            f"(start: {self.start_unix}, end: {self.end_unix})"
        )
        print(line, flush=True)
        if self.logfile is not None:
            with open(self.logfile, "a") as fh:
                fh.write(line + "\n")


class TimeStats(BaseModel):
    tags: list[str]
    elapsed_s: float
    start_unix: float
    end_unix: float


def parse_timer_logs(text: str) -> list[TimeStats]:
    """Parse ``[timer]`` lines back into structured stats."""
    out: list[TimeStats] = []
    for m in _TIMER_RE.finditer(text):
        tags = [t.strip() for t in m.group("tags").split(",") if t.strip()]
        out.append(
            TimeStats(
                tags=tags,
                elapsed_s=float(m.group("elapsed")),
                start_unix=float(m.group("start")),
                end_unix=float(m.group("end")),
            )
        )
    return out


def retry(tries: int = 5, base_delay: float = 0.5, max_delay: float = 30.0,
          exceptions: tuple[type[BaseException], ...] = (Exception,)):
    """Tiny exponential-backoff retry decorator (replaces the `backoff` dep
    used by the reference MCQA harness, mcqa/*_v3.py:1956-1963)."""

    def deco(fn):
        import functools

        @functools.wraps(fn)
        def wrapped(*args: Any, **kwargs: Any):
            delay = base_delay
            for attempt in range(tries):
                try:
                    return fn(*args, **kwargs)
                except exceptions:
                    if attempt == tries - 1:
                        raise
                    time.sleep(delay)
                    delay = min(delay * 2, max_delay)
        return wrapped

    return deco
