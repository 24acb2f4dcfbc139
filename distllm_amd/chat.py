"""Terminal RAG chat application.

Reference parity: distllm/chat.py (ConversationPromptTemplate carrying
history + a retrieval-context block, RagGenerator with retrieval debug
dump, /inspect command, top_k=20 / threshold=0.1 defaults, transcript
saved to a timestamped file) and distllm/chat_argoproxy.py's pluggable
generator backends (here: native in-process engine or any
OpenAI-compatible HTTP endpoint).
"""

from __future__ import annotations

import os
import re
import time
from pathlib import Path
from typing import Any


from .generate import get_generator
from .rag.response_synthesizer import RagGenerator
from .rag.search import RetrieverConfig
from .utils import BaseConfig

_ENV_RE = re.compile(r"\$\{env:([A-Za-z_][A-Za-z0-9_]*)\}")


def substitute_env(obj: Any) -> Any:
    """`${env:VAR}` substitution in config values (reference
    chat_argoproxy.py:510-549)."""
    if isinstance(obj, str):
        return _ENV_RE.sub(lambda m: os.environ.get(m.group(1), ""), obj)
    if isinstance(obj, dict):
        return {k: substitute_env(v) for k, v in obj.items()}
    if isinstance(obj, list):
        return [substitute_env(v) for v in obj]
    return obj


class ConversationPromptTemplate:
    """History-aware prompt with a retrieval-context block."""

    def __init__(self, system_prompt: str = ""):
        self.system_prompt = system_prompt or (
            "You are a helpful scientific assistant. Use the retrieved "
            "context when it is relevant and say so when it is not."
        )
        self.history: list[tuple[str, str]] = []  # (role, content)

    def preprocess(self, text, contexts=None, scores=None) -> list[str]:
        if isinstance(text, list):
            text = text[0]
        parts = [self.system_prompt, ""]
        for role, content in self.history:
            parts.append(f"{role.capitalize()}: {content}")
        if contexts:
            ctx = contexts[0]
            sc = scores[0] if scores else [0.0] * len(ctx)
            parts.append("[Context from retrieval]")
            for c, s in zip(ctx, sc):
                parts.append(f"(score {s:.3f}) {c}")
            parts.append("[End context]")
        parts.append(f"User: {text}")
        parts.append("Assistant:")
        return ["\n".join(parts)]

    def postprocess(self, responses: list[str]) -> list[str]:
        return [r.strip() for r in responses]


def translate_target_config(gen: dict[str, Any]) -> dict[str, Any]:
    """Accept the reference chat_argoproxy `_target_` class-dispatch
    YAMLs (reference distllm/chat_argoproxy.py:510-549) by translating
    the three generator config classes onto our generator names:

      VLLMGeneratorConfig    {server, port, api_key, model, ...}
      ArgoGeneratorConfig    {model, base_url, api_key, ...}
      OpenAIAPIGeneratorConfig {model, api_key, base_url?, ...}

    all map to the OpenAI-compatible HTTP client. Bare
    server/port-shaped dicts (the reference chat.py
    VLLMGeneratorSettings blocks in examples/chat/*.yaml) translate the
    same way; dicts already naming one of our backends pass through
    unchanged."""
    if "_target_" not in gen:
        if "name" not in gen and "server" in gen and "port" in gen:
            gen = {**gen, "_target_": "VLLMGeneratorConfig"}
        else:
            return gen
    gen = dict(gen)
    target = gen.pop("_target_")
    common = {
        k: gen[k] for k in ("model", "temperature", "max_tokens")
        if k in gen
    }
    if target == "VLLMGeneratorConfig":
        out = {"name": "openai_http",
               "base_url": f"http://{gen['server']}:{gen['port']}/v1",
               "api_key": gen.get("api_key", ""), **common}
    elif target == "ArgoGeneratorConfig":
        out = {"name": "openai_http",
               "base_url": gen.get("base_url", "http://localhost:56267"),
               "api_key": gen.get("api_key", ""), **common}
    elif target == "OpenAIAPIGeneratorConfig":
        out = {"name": "openai_http",
               "base_url": gen.get("base_url")
               or "https://api.openai.com/v1",
               "api_key": gen.get("api_key", ""), **common}
    else:
        raise ValueError(
            f"Unknown generator config class: {target}. Available: "
            "['VLLMGeneratorConfig', 'ArgoGeneratorConfig', "
            "'OpenAIAPIGeneratorConfig']")
    return out


class ChatAppConfig(BaseConfig):
    generator_config: dict[str, Any]
    retriever_config: RetrieverConfig | None = None
    system_prompt: str = ""
    retrieval_top_k: int = 20
    retrieval_score_threshold: float = 0.1
    transcript_dir: Path = Path("chat_transcripts")

    @classmethod
    def from_yaml(cls, path):
        import yaml

        with open(path) as fh:
            data = yaml.safe_load(fh)
        data = substitute_env(data or {})
        # reference chat YAML dialect (examples/chat/*.yaml): nested
        # rag_configs block + save_conversation_path
        if isinstance(data.get("rag_configs"), dict):
            rc = data.pop("rag_configs")
            data.setdefault("generator_config",
                            rc.get("generator_config", {}))
            if rc.get("retriever_config") is not None:
                data.setdefault("retriever_config",
                                rc["retriever_config"])
        if "save_conversation_path" in data:
            data.setdefault("transcript_dir",
                            data.pop("save_conversation_path"))
        if isinstance(data.get("generator_config"), dict):
            data["generator_config"] = translate_target_config(
                data["generator_config"])
        return cls(**data)


class ChatSession:
    def __init__(self, config: ChatAppConfig, batching: bool = False):
        self.config = config
        generator = get_generator(substitute_env(
            dict(config.generator_config)))
        retriever = (config.retriever_config.get_retriever()
                     if config.retriever_config is not None else None)
        self.rag = RagGenerator(generator=generator, retriever=retriever)
        self.template = ConversationPromptTemplate(config.system_prompt)
        self.last_retrieval: list[tuple[float, str]] = []
        self.batcher = None
        if batching:
            if hasattr(generator, "generate_one"):
                # native engine: request-level continuous batching
                generator.get_loop()
            else:
                from .engine.batcher import DynamicBatcher

                self.batcher = DynamicBatcher(generator)

    def _generate(self, prompt: str) -> str:
        gen = self.rag.generator
        if hasattr(gen, "generate_one") and getattr(gen, "_loop", None):
            return gen.generate_one(prompt)
        if self.batcher is not None:
            return self.batcher.generate_one(prompt)
        return gen.generate([prompt])[0]

    def answer(self, history: list[tuple[str, str]], user_text: str
               ) -> tuple[str, list[tuple[float, str]]]:
        """Stateless turn: caller supplies history; returns (answer,
        retrieval_dump). Thread-safe (used by the server per request)."""
        contexts = scores = None
        retrieval: list[tuple[float, str]] = []
        if self.rag.retriever is not None:
            results, _ = self.rag.retriever.search(
                [user_text], top_k=self.config.retrieval_top_k,
                score_threshold=self.config.retrieval_score_threshold,
            )
            idxs = results.total_indices[0]
            texts = self.rag.retriever.get_texts(idxs)
            contexts = [texts]
            scores = [results.total_scores[0]]
            retrieval = list(zip(results.total_scores[0], texts))
        tmpl = ConversationPromptTemplate(self.config.system_prompt)
        tmpl.history = list(history)
        prompt = tmpl.preprocess(user_text, contexts, scores)[0]
        answer = tmpl.postprocess([self._generate(prompt)])[0]
        return answer, retrieval

    def turn(self, user_text: str) -> str:
        answer, retrieval = self.answer(self.template.history, user_text)
        self.last_retrieval = retrieval
        self.template.history.append(("user", user_text))
        self.template.history.append(("assistant", answer))
        return answer

    def inspect(self) -> str:
        if not self.last_retrieval:
            return "(no retrieval yet)"
        lines = [f"[{s:.4f}] {t[:200]}" for s, t in self.last_retrieval]
        return "\n".join(lines)

    def save_transcript(self) -> Path | None:
        if not self.template.history:
            return None
        d = Path(self.config.transcript_dir)
        d.mkdir(parents=True, exist_ok=True)
        out = d / f"chat_{time.strftime('%Y%m%d_%H%M%S')}.txt"
        with open(out, "w") as fh:
            for role, content in self.template.history:
                fh.write(f"{role}: {content}\n\n")
        return out


def chat_with_model(config: ChatAppConfig) -> None:
    session = ChatSession(config)
    print("distllm-amd chat — /inspect shows last retrieval, /quit exits")
    try:
        while True:
            try:
                user = input("You: ").strip()
            except EOFError:
                break
            if not user:
                continue
            if user in ("/quit", "/exit"):
                break
            if user == "/inspect":
                print(session.inspect())
                continue
            print("Assistant:", session.turn(user))
    finally:
        out = session.save_transcript()
        if out:
            print(f"transcript saved to {out}")
