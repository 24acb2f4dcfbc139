"""OpenAI-compatible FastAPI chat server over the RAG pipeline.

Reference parity: distllm/chat_server.py:42-270 — POST
/v1/chat/completions (+/health), env config DISTLLM_CHAT_CONFIG,
DISTLLM_TOP_K / DISTLLM_SCORE_THRESHOLD env defaults, single-delta SSE
streaming shim, threadpool offload of the sync generate.
"""

from __future__ import annotations

import json
import os
import time
import uuid
from typing import Any

from fastapi import FastAPI, HTTPException
from fastapi.concurrency import run_in_threadpool
from fastapi.responses import StreamingResponse
from pydantic import BaseModel, Field

app = FastAPI(title="distllm-amd chat server")

_session = None


def _get_session():
    global _session
    if _session is None:
        from .chat import ChatAppConfig, ChatSession

        cfg_path = os.environ.get("DISTLLM_CHAT_CONFIG")
        if not cfg_path:
            raise HTTPException(500, "DISTLLM_CHAT_CONFIG not set")
        config = ChatAppConfig.from_yaml(cfg_path)
        if os.environ.get("DISTLLM_TOP_K"):
            config.retrieval_top_k = int(os.environ["DISTLLM_TOP_K"])
        if os.environ.get("DISTLLM_SCORE_THRESHOLD"):
            config.retrieval_score_threshold = float(
                os.environ["DISTLLM_SCORE_THRESHOLD"]
            )
        batching = os.environ.get("DISTLLM_BATCHING", "1") != "0"
        _session = ChatSession(config, batching=batching)
    return _session


class ChatMessage(BaseModel):
    role: str
    content: str


class ChatCompletionRequest(BaseModel):
    model: str = "distllm-amd"
    messages: list[ChatMessage]
    temperature: float | None = None
    max_tokens: int | None = None
    stream: bool = False
    extra: dict[str, Any] = Field(default_factory=dict)


@app.get("/health")
def health() -> dict:
    return {"status": "ok"}


@app.get("/v1/models")
def models() -> dict:
    return {"object": "list",
            "data": [{"id": "distllm-amd", "object": "model"}]}


@app.post("/v1/chat/completions")
async def chat_completions(req: ChatCompletionRequest):
    session = _get_session()
    user_msgs = [m.content for m in req.messages if m.role == "user"]
    if not user_msgs:
        raise HTTPException(400, "no user message")
    # stateless per request: client-supplied history, concurrent
    # requests batch into one engine call (DynamicBatcher)
    history = [(m.role, m.content) for m in req.messages[:-1]
               if m.role in ("user", "assistant")]
    answer, _retrieval = await run_in_threadpool(
        session.answer, history, user_msgs[-1])
    rid = f"chatcmpl-{uuid.uuid4().hex[:16]}"
    created = int(time.time())
    if req.stream:
        def sse():
            chunk = {
                "id": rid, "object": "chat.completion.chunk",
                "created": created, "model": req.model,
                "choices": [{"index": 0,
                             "delta": {"role": "assistant",
                                       "content": answer},
                             "finish_reason": None}],
            }
            yield f"data: {json.dumps(chunk)}\n\n"
            done = {
                "id": rid, "object": "chat.completion.chunk",
                "created": created, "model": req.model,
                "choices": [{"index": 0, "delta": {},
                             "finish_reason": "stop"}],
            }
            yield f"data: {json.dumps(done)}\n\n"
            yield "data: [DONE]\n\n"

        return StreamingResponse(sse(), media_type="text/event-stream")
    return {
        "id": rid,
        "object": "chat.completion",
        "created": created,
        "model": req.model,
        "choices": [{
            "index": 0,
            "message": {"role": "assistant", "content": answer},
            "finish_reason": "stop",
        }],
        "usage": {"prompt_tokens": 0, "completion_tokens": 0,
                  "total_tokens": 0},
    }
