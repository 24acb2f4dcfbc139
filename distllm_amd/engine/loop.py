"""Online engine loop: vLLM-style request-level continuous batching.

Reference parity: the reference delegated this to vLLM's AsyncLLMEngine
behind an OpenAI endpoint (distllm mcqa launches `vllm serve`,
mcqa/rag_argonium_score_parallel_v3.py:1206-1354; client batching in
the MCQA queue, v3:984-989,1419-1449). Here the engine loop is native.

DynamicBatcher groups concurrent requests into discrete generate()
calls, which still head-of-line blocks a second batch behind the first.
EngineLoop owns the engine on a background thread and admits requests
into the RUNNING decode batch at token granularity: a request arriving
mid-generation is prefilled at the next step boundary and decodes
alongside the others.
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field

from .engine import LLMEngine, SamplingParams, _Seq


@dataclass
class _Request:
    prompt_ids: list[int]
    max_tokens: int
    event: threading.Event = field(default_factory=threading.Event)
    result: list[int] | None = None
    error: Exception | None = None
    seq: _Seq | None = None


class EngineLoop:
    def __init__(self, engine: LLMEngine, params: SamplingParams):
        self.engine = engine
        self.params = params
        self._lock = threading.Lock()
        self._cv = threading.Condition(self._lock)
        self._waiting: list[_Request] = []
        self._running: list[_Request] = []
        self._closed = False
        # pre-capture every pow2 decode-graph bucket BEFORE serving:
        # capturing graphs mid-flight (batch bucket shrinks as requests
        # drain) while submitter threads run was observed to
        # intermittently abort the HIP runtime under stress
        if engine.use_graphs:
            buckets, b = [], 1
            while b < engine.max_batch_size:
                buckets.append(b)
                b <<= 1
            buckets.append(engine.max_batch_size)
            engine.warm_graphs(sorted(set(buckets)))
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def submit(self, prompt_ids: list[int],
               max_tokens: int | None = None) -> _Request:
        req = _Request(prompt_ids=list(prompt_ids),
                       max_tokens=max_tokens or self.params.max_tokens)
        with self._cv:
            if self._closed:
                raise RuntimeError("engine loop closed")
            self._waiting.append(req)
            self._cv.notify()
        return req

    def generate(self, prompts_ids: list[list[int]],
                 max_tokens: int | None = None,
                 timeout: float = 600.0) -> list[list[int]]:
        reqs = [self.submit(p, max_tokens) for p in prompts_ids]
        outs = []
        for r in reqs:
            if not r.event.wait(timeout):
                raise TimeoutError("generation timed out")
            if r.error is not None:
                raise r.error
            outs.append(r.result or [])
        return outs

    # ---- background loop ------------------------------------------------
    def _admit(self) -> list[_Request]:
        eng = self.engine
        admitted: list[_Request] = []
        with self._cv:
            while (self._waiting
                   and len(self._running) + len(admitted)
                   < eng.max_batch_size):
                req = self._waiting[0]
                plen = min(len(req.prompt_ids), eng.max_model_len - 1)
                need = eng.blockman.blocks_for(
                    min(plen + req.max_tokens, eng.max_model_len))
                if not eng.blockman.can_alloc(need):
                    break
                self._waiting.pop(0)
                seq = _Seq(seq_id=id(req),
                           prompt_ids=req.prompt_ids[-plen:],
                           max_tokens=req.max_tokens)
                seq.blocks = eng.blockman.alloc(need)
                req.seq = seq
                admitted.append(req)
        return admitted

    def _finish(self, req: _Request, out: list[int] | None,
                err: Exception | None = None) -> None:
        req.result = out
        req.error = err
        req.event.set()

    def _loop(self) -> None:
        eng = self.engine
        params = self.params
        results: dict[int, list[int]] = {}
        while True:
            admitted = self._admit()
            if admitted:
                try:
                    seqs = [r.seq for r in admitted]
                    eng._run_prefill(seqs, params, results)
                except Exception as e:
                    for r in admitted:
                        eng.blockman.release(r.seq.blocks)
                        self._finish(r, None, e)
                    admitted = []
                self._running.extend(r for r in admitted)
            # harvest + decode
            still: list[_Request] = []
            for r in self._running:
                done = (r.seq.done
                        or len(r.seq.out_ids) >= r.max_tokens)
                if done:
                    out = results.pop(r.seq.seq_id, r.seq.out_ids)
                    if not r.seq.done:  # hit per-request cap
                        eng.blockman.release(r.seq.blocks)
                        r.seq.blocks = []
                    self._finish(r, out)
                else:
                    still.append(r)
            self._running = still
            if not self._running:
                with self._cv:
                    if self._closed:
                        return
                    if not self._waiting:
                        self._cv.wait(0.05)
                continue
            try:
                eng._run_decode([r.seq for r in self._running], params,
                                results)
            except Exception as e:
                for r in self._running:
                    eng.blockman.release(r.seq.blocks)
                    self._finish(r, None, e)
                self._running = []

    def close(self) -> None:
        with self._cv:
            self._closed = True
            self._cv.notify_all()
        self._thread.join(timeout=5)
