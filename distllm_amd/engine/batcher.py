"""Dynamic request batcher: groups concurrent generate() calls into one
engine batch (the serving-side analog of the reference MCQA harness's
client-side request batching, mcqa/*_v3.py:1407-1605 — theirs batched
HTTP calls client-side; here the server batches into the continuous-
batching engine directly).
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field


@dataclass
class _Pending:
    prompt: str
    event: threading.Event = field(default_factory=threading.Event)
    result: str | None = None
    error: Exception | None = None


class DynamicBatcher:
    def __init__(self, generator, max_batch: int = 32,
                 window_ms: float = 10.0):
        self.generator = generator
        self.max_batch = max_batch
        self.window_s = window_ms / 1000.0
        self._lock = threading.Lock()
        self._cv = threading.Condition(self._lock)
        self._queue: list[_Pending] = []
        self._closed = False
        self._worker = threading.Thread(target=self._loop, daemon=True)
        self._worker.start()

    def generate_one(self, prompt: str, timeout: float = 600.0) -> str:
        p = _Pending(prompt=prompt)
        with self._cv:
            self._queue.append(p)
            self._cv.notify()
        if not p.event.wait(timeout):
            raise TimeoutError("generation timed out")
        if p.error is not None:
            raise p.error
        return p.result or ""

    def _loop(self) -> None:
        while True:
            with self._cv:
                while not self._queue and not self._closed:
                    self._cv.wait()
                if self._closed:
                    return
                # small window to let concurrent requests join the batch
                self._cv.wait(self.window_s)
                batch = self._queue[: self.max_batch]
                del self._queue[: len(batch)]
            try:
                outs = self.generator.generate([b.prompt for b in batch])
                for b, o in zip(batch, outs):
                    b.result = o
            except Exception as e:  # propagate to every waiter
                for b in batch:
                    b.error = e
            for b in batch:
                b.event.set()

    def close(self) -> None:
        with self._cv:
            self._closed = True
            self._cv.notify_all()
