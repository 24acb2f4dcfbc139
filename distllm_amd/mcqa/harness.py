"""MCQA benchmarking harness: large-scale multiple-choice QA scoring
with an LLM grader, retrieval provenance tracking and checkpoint/resume.

Capability parity with reference
distllm/mcqa/rag_argonium_score_parallel_v3.py (3786 LoC):
  - MCQAConfig tree {model, rag, processing, output} (v3:401-444)
  - chunk-ID provenance sha256(path)[:16]_idx (v3:447-478)
  - source-chunk + question-hash retrieval tracking (v3:504,594)
  - self-booting local server subprocess with port autodiscovery
    (v3:1002-1107) — boots OUR OpenAI-compatible chat server
  - RagGenerator-with-chunk-logging returning (response, retrieval_info)
  - LLM grader with an escalating JSON-parse retry ladder (v3:2065-2228)
  - ThreadPool parallel question processing (v3:2245)
  - checkpoint save / resume / auto-find (v3:2891-3073)
  - metrics: overall/MC/QA accuracy, source-retrieval rate,
    question-hash retrieval rate (v3:3600-3700)
"""

from __future__ import annotations

import hashlib
import json
import re
import socket
import subprocess
import sys
import time
from concurrent.futures import ThreadPoolExecutor, as_completed
from pathlib import Path
from typing import Any

from ..utils import BaseConfig, retry

# ---------------------------------------------------------------------------
# config tree
# ---------------------------------------------------------------------------


class ModelConfig(BaseConfig):
    """Answering model: an OpenAI-compatible endpoint, optionally booted
    locally (boot_local -> we spawn our own chat/engine server)."""

    base_url: str = ""
    model: str = "distllm-amd"
    api_key_env: str = "OPENAI_API_KEY"
    api_key: str = ""  # literal key (reference v2 generator_settings)
    temperature: float = 0.0
    max_tokens: int = 512
    boot_local: bool = False
    local_config_path: Path | None = None  # ChatAppConfig yaml
    port: int = 0  # 0 -> auto-discover a free port

    def model_post_init(self, _ctx) -> None:
        if self.boot_local and self.local_config_path is None:
            raise ValueError(
                "boot_local=True requires local_config_path"
            )


class GraderConfig(BaseConfig):
    base_url: str = ""
    model: str = "distllm-amd"
    api_key_env: str = "OPENAI_API_KEY"
    api_key: str = ""
    temperature: float = 0.0
    max_tokens: int = 256


class RagSettings(BaseConfig):
    use_rag: bool = False
    retriever_config: dict[str, Any] | None = None
    top_k: int = 5
    score_threshold: float = 0.0


class ProcessingConfig(BaseConfig):
    parallel_workers: int = 4
    checkpoint_interval: int = 10
    save_incremental: bool = False
    random_selection: int = 0  # 0 = all questions
    seed: int = 42


class OutputConfig(BaseConfig):
    results_file: Path = Path("mcqa_results.json")
    checkpoint_dir: Path = Path("mcqa_checkpoints")


def _load_model_server(shortname: str, config_file: str | Path) -> dict:
    """Resolve a grader/model shortname from an argonium-style
    model_servers.yaml (reference v2:603-634: servers: [{shortname,
    openai_api_base, openai_api_key, openai_model}, ...])."""
    import yaml as _yaml

    data = _yaml.safe_load(Path(config_file).read_text()) or {}
    for server in data.get("servers", []):
        if server.get("shortname") == shortname:
            return server
    avail = [s.get("shortname") for s in data.get("servers", [])]
    raise ValueError(f"model {shortname!r} not found in "
                     f"{config_file}; available: {avail}")


def translate_v2_config(raw: dict, base_dir: Path | None = None) -> dict:
    """Map the reference MCQA v2/v3 YAML tree (reference
    rag_argonium_score_parallel_v2.py config schema: nested
    model.generator/generator_settings, rag.enabled/retrieval_top_k,
    processing.*, output.output_directory+output_prefix) onto
    MCQAConfig's tree. Native-shaped dicts pass through untouched."""
    model = raw.get("model")
    if not (isinstance(model, dict)
            and ("generator" in model or "generator_settings" in model)):
        return raw
    gs = model.get("generator_settings", {}) or {}
    base_url = gs.get("base_url", "")
    if not base_url and gs.get("server"):
        base_url = f"http://{gs['server']}:{gs.get('port', 8000)}/v1"
    out: dict = {"questions_file": raw.get("questions_file", "")}
    out["model"] = {
        "base_url": base_url,
        "model": gs.get("model", "distllm-amd"),
        "api_key": str(gs.get("api_key", "") or ""),
        "temperature": gs.get("temperature", 0.0),
        "max_tokens": gs.get("max_tokens", 512),
    }
    grader: dict = {}
    if model.get("grader_shortname"):
        cf = Path(model.get("model_config_file", "model_servers.yaml"))
        if not cf.is_absolute() and base_dir is not None:
            cf = base_dir / cf
        if cf.exists():
            srv = _load_model_server(model["grader_shortname"], cf)
            grader = {
                "base_url": srv.get("openai_api_base", ""),
                "model": srv.get("openai_model",
                                 model["grader_shortname"]),
                "api_key": str(srv.get("openai_api_key", "") or ""),
            }
        else:
            grader = {"model": model["grader_shortname"]}
    out["grader"] = grader or dict(out["model"])
    rag = raw.get("rag", {}) or {}
    retr = rag.get("retriever_config")
    if retr is None and rag.get("rag_config_file"):
        import yaml as _yaml

        rf = Path(rag["rag_config_file"])
        if not rf.is_absolute() and base_dir is not None:
            rf = base_dir / rf
        if rf.exists():
            retr = _yaml.safe_load(rf.read_text())
    out["rag"] = {
        "use_rag": bool(rag.get("enabled", False)),
        "retriever_config": retr,
        "top_k": rag.get("retrieval_top_k", 5),
        "score_threshold": rag.get("retrieval_score_threshold", 0.0),
    }
    proc = raw.get("processing", {}) or {}
    out["processing"] = {
        "parallel_workers": proc.get("parallel_workers", 4),
        "random_selection": proc.get("random_selection") or 0,
        "seed": proc.get("random_seed") or 42,
    }
    o = raw.get("output", {}) or {}
    outdir = Path(o.get("output_directory", "."))
    prefix = o.get("output_prefix", "mcqa")
    out["output"] = {
        "results_file": outdir / f"{prefix}_results.json",
        "checkpoint_dir": outdir / f"{prefix}_checkpoints",
    }
    return out


class MCQAConfig(BaseConfig):
    questions_file: Path
    model: ModelConfig
    grader: GraderConfig
    rag: RagSettings = RagSettings()
    processing: ProcessingConfig = ProcessingConfig()
    output: OutputConfig = OutputConfig()

    @classmethod
    def from_yaml(cls, path) -> "MCQAConfig":
        import yaml as _yaml

        raw = _yaml.safe_load(Path(path).read_text()) or {}
        raw = translate_v2_config(raw, base_dir=Path(path).parent)
        return cls(**raw)


# ---------------------------------------------------------------------------
# provenance helpers
# ---------------------------------------------------------------------------


def make_chunk_id(path: str, idx: int) -> str:
    """sha256(path)[:16]_idx (reference v3:447-478)."""
    return f"{hashlib.sha256(str(path).encode()).hexdigest()[:16]}_{idx}"


def question_hash(question: str) -> str:
    return hashlib.sha256(question.strip().encode()).hexdigest()[:16]


# ---------------------------------------------------------------------------
# local server boot (reference v3:1002-1405)
# ---------------------------------------------------------------------------


def find_free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class LocalServer:
    """Boots our OpenAI-compatible chat server as a subprocess and waits
    for /health; collects startup diagnostics on failure."""

    def __init__(self, config_path: Path, port: int = 0,
                 startup_timeout: float = 600.0):
        self.port = port or find_free_port()
        self.config_path = config_path
        self.startup_timeout = startup_timeout
        self.proc: subprocess.Popen | None = None

    @property
    def base_url(self) -> str:
        return f"http://127.0.0.1:{self.port}/v1"

    def start(self) -> None:
        import os

        env = dict(os.environ, DISTLLM_CHAT_CONFIG=str(self.config_path))
        self.proc = subprocess.Popen(
            [sys.executable, "-m", "uvicorn",
             "distllm_amd.chat_server:app", "--host", "127.0.0.1",
             "--port", str(self.port)],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        )
        import httpx

        t0 = time.time()
        url = f"http://127.0.0.1:{self.port}/health"
        while time.time() - t0 < self.startup_timeout:
            if self.proc.poll() is not None:
                out = (self.proc.stdout.read() or b"").decode(
                    errors="replace")
                raise RuntimeError(
                    f"local server exited rc={self.proc.returncode}; "
                    f"log tail:\n{out[-2000:]}"
                )
            try:
                if httpx.get(url, timeout=2.0).status_code == 200:
                    return
            except Exception:
                pass
            time.sleep(1.0)
        self.stop()
        raise TimeoutError("local server failed to become healthy")

    def stop(self) -> None:
        """Graceful-first shutdown: SIGINT lets uvicorn close and the
        process destroy its GPU context cleanly — hard-killing a
        process mid-kernel can leave the device queues in a state that
        intermittently aborts LATER GPU work in this process (observed
        as full-test-suite flakes surfacing tests later)."""
        if self.proc is not None and self.proc.poll() is None:
            import signal

            self.proc.send_signal(signal.SIGINT)
            try:
                self.proc.wait(timeout=15)
            except subprocess.TimeoutExpired:
                self.proc.terminate()
                try:
                    self.proc.wait(timeout=10)
                except subprocess.TimeoutExpired:
                    self.proc.kill()
        self.proc = None


# ---------------------------------------------------------------------------
# clients
# ---------------------------------------------------------------------------


class _ChatClient:
    def __init__(self, base_url: str, model: str, api_key_env: str,
                 temperature: float, max_tokens: int,
                 api_key: str = ""):
        import os

        import httpx

        headers = {}
        key = api_key or os.environ.get(api_key_env, "")
        if key:
            headers["Authorization"] = f"Bearer {key}"
        self.client = httpx.Client(base_url=base_url, headers=headers,
                                   timeout=300.0)
        self.model = model
        self.temperature = temperature
        self.max_tokens = max_tokens

    @retry(tries=5, base_delay=1.0, max_delay=60.0)
    def complete(self, prompt: str, temperature: float | None = None) -> str:
        r = self.client.post("/chat/completions", json={
            "model": self.model,
            "messages": [{"role": "user", "content": prompt}],
            "temperature": (self.temperature if temperature is None
                            else temperature),
            "max_tokens": self.max_tokens,
        })
        r.raise_for_status()
        return r.json()["choices"][0]["message"]["content"]


# ---------------------------------------------------------------------------
# grading (reference v3:2017-2243 retry ladder)
# ---------------------------------------------------------------------------

_GRADE_PROMPTS = [
    # attempt 1: friendly
    ("Grade the following answer to a question against the reference "
     "answer.\nQuestion: {question}\nReference answer: {reference}\n"
     "Model answer: {answer}\n"
     'Respond with JSON: {{"correct": true/false, "reason": "..."}}'),
    # attempt 2: stricter
    ("You are a strict grader. Output ONLY a JSON object, nothing else."
     '\nSchema: {{"correct": <true|false>, "reason": "<short>"}}\n'
     "Question: {question}\nReference answer: {reference}\n"
     "Model answer: {answer}\nJSON:"),
    # attempt 3: minimal
    ('Answer with exactly {{"correct": true}} or {{"correct": false}}.\n'
     "Question: {question}\nReference: {reference}\nAnswer: {answer}\n"),
]

_JSON_RE = re.compile(r"\{.*?\}", re.DOTALL)


def parse_grade_json(text: str) -> dict | None:
    for m in _JSON_RE.finditer(text):
        try:
            obj = json.loads(m.group(0))
            if isinstance(obj, dict) and "correct" in obj:
                return obj
        except json.JSONDecodeError:
            continue
    # last resort: keyword scan
    low = text.lower()
    if '"correct": true' in low or "correct: true" in low:
        return {"correct": True, "reason": "keyword"}
    if '"correct": false' in low or "correct: false" in low:
        return {"correct": False, "reason": "keyword"}
    return None


def grade_answer(grader: _ChatClient, question: str, reference: str,
                 answer: str) -> dict:
    for attempt, template in enumerate(_GRADE_PROMPTS):
        prompt = template.format(question=question, reference=reference,
                                 answer=answer)
        try:
            resp = grader.complete(prompt)
        except Exception as e:
            if attempt == len(_GRADE_PROMPTS) - 1:
                return {"correct": False,
                        "reason": f"grader error: {e}",
                        "grader_failed": True}
            continue
        parsed = parse_grade_json(resp)
        if parsed is not None:
            parsed["attempts"] = attempt + 1
            return parsed
    return {"correct": False, "reason": "unparseable grader output",
            "grader_failed": True}


# ---------------------------------------------------------------------------
# checkpointing (reference v3:2891-3073)
# ---------------------------------------------------------------------------


class Checkpointer:
    VERSION = 1

    def __init__(self, ckpt_dir: Path, questions_file: Path, model: str):
        self.dir = Path(ckpt_dir)
        self.dir.mkdir(parents=True, exist_ok=True)
        stem = Path(questions_file).stem
        safe_model = re.sub(r"[^A-Za-z0-9_.-]", "_", model)
        self.pattern = f"ckpt_{stem}_{safe_model}_"

    def save(self, completed: dict[int, dict], config_dump: dict) -> Path:
        payload = {
            "version": self.VERSION,
            "timestamp": time.time(),
            "completed_indices": sorted(completed),
            "results": {str(k): v for k, v in completed.items()},
            "config": config_dump,
        }
        out = self.dir / f"{self.pattern}{int(time.time())}.json"
        tmp = out.with_suffix(".tmp")
        tmp.write_text(json.dumps(payload))
        tmp.rename(out)
        return out

    def find_latest(self) -> Path | None:
        cands = sorted(self.dir.glob(f"{self.pattern}*.json"))
        return cands[-1] if cands else None

    def load(self, path: Path, config_dump: dict) -> dict[int, dict]:
        data = json.loads(Path(path).read_text())
        if data.get("version") != self.VERSION:
            print(f"[ckpt] version mismatch in {path}, ignoring")
            return {}
        if data.get("config", {}).get("questions_file") != \
                config_dump.get("questions_file"):
            print(f"[ckpt] different questions file in {path}, ignoring")
            return {}
        return {int(k): v for k, v in data.get("results", {}).items()}


# ---------------------------------------------------------------------------
# the harness
# ---------------------------------------------------------------------------


def _load_questions(path: Path) -> list[dict]:
    text = Path(path).read_text()
    data = json.loads(text)
    if isinstance(data, dict):
        data = [e for v in data.values()
                for e in (v if isinstance(v, list) else [v])]
    return data


def _answer_question(client: _ChatClient, retriever, rag_cfg: RagSettings,
                     q: dict) -> tuple[str, dict]:
    """Answer with optional retrieval; returns (answer, retrieval_info)
    with chunk-ID + hash provenance (the RagGeneratorWithChunkLogging
    role, reference v3:1744-1910)."""
    question = q.get("question", "")
    retrieval_info: dict = {"used_rag": False}
    prompt = question
    if rag_cfg.use_rag and retriever is not None:
        results, _ = retriever.search([question], top_k=rag_cfg.top_k,
                                      score_threshold=rag_cfg.score_threshold)
        idxs = results.total_indices[0]
        texts = retriever.get_texts(idxs)
        try:
            paths = retriever.get(idxs, "path")
        except (KeyError, Exception):
            paths = [""] * len(idxs)
        chunk_ids = [make_chunk_id(p, i) for p, i in zip(paths, idxs)]
        retrieval_info = {
            "used_rag": True,
            "retrieved_indices": [int(i) for i in idxs],
            "retrieved_chunk_ids": chunk_ids,
            "scores": results.total_scores[0],
            "source_chunk_retrieved": (
                q.get("source_chunk_id") in chunk_ids
                if q.get("source_chunk_id") else None
            ),
            "question_hash_retrieved": any(
                question_hash(question) in (t or "") for t in texts
            ),
        }
        ctx = "\n".join(f"Context: {t}" for t in texts)
        prompt = f"{ctx}\n\nQuestion: {question}\nAnswer:"
    answer = client.complete(prompt)
    return answer, retrieval_info


def run_mcqa(config: MCQAConfig) -> dict:
    questions = _load_questions(config.questions_file)
    proc = config.processing
    if proc.random_selection and proc.random_selection < len(questions):
        import random as _r

        rng = _r.Random(proc.seed)
        questions = rng.sample(questions, proc.random_selection)

    server: LocalServer | None = None
    base_url = config.model.base_url
    if config.model.boot_local:
        server = LocalServer(config.model.local_config_path,
                             config.model.port)
        server.start()
        base_url = server.base_url

    try:
        client = _ChatClient(base_url, config.model.model,
                             config.model.api_key_env,
                             config.model.temperature,
                             config.model.max_tokens,
                             api_key=config.model.api_key)
        grader = _ChatClient(config.grader.base_url or base_url,
                             config.grader.model,
                             config.grader.api_key_env,
                             config.grader.temperature,
                             config.grader.max_tokens,
                             api_key=config.grader.api_key)
        retriever = None
        if config.rag.use_rag and config.rag.retriever_config:
            from ..rag.search import RetrieverConfig

            retriever = RetrieverConfig(
                **config.rag.retriever_config).get_retriever()

        ckpt = Checkpointer(config.output.checkpoint_dir,
                            config.questions_file, config.model.model)
        cfg_dump = json.loads(config.model_dump_json())
        completed: dict[int, dict] = {}
        latest = ckpt.find_latest()
        if latest is not None:
            completed = ckpt.load(latest, cfg_dump)
            print(f"[ckpt] resumed {len(completed)} results from {latest}")

        todo = [i for i in range(len(questions)) if i not in completed]

        def work(i: int) -> tuple[int, dict]:
            q = questions[i]
            t0 = time.time()
            answer, rinfo = _answer_question(client, retriever, config.rag,
                                             q)
            ref = str(q.get("answer", q.get("ideal", "")))
            grade = grade_answer(grader, q.get("question", ""), ref, answer)
            return i, {
                "question": q.get("question", ""),
                "reference": ref,
                "answer": answer,
                "grade": grade,
                "type": q.get("type", "mc" if "options" in q else "qa"),
                "retrieval": rinfo,
                "elapsed_s": time.time() - t0,
            }

        done_since_save = 0
        with ThreadPoolExecutor(max_workers=proc.parallel_workers) as pool:
            futures = {pool.submit(work, i): i for i in todo}
            for fut in as_completed(futures):
                i, res = fut.result()
                completed[i] = res
                done_since_save += 1
                if (proc.save_incremental
                        or done_since_save >= proc.checkpoint_interval):
                    ckpt.save(completed, cfg_dump)
                    done_since_save = 0
                n = len(completed)
                if n % 10 == 0 or n == len(questions):
                    print(f"[mcqa] {n}/{len(questions)} done")

        ckpt.save(completed, cfg_dump)
        results = [completed[i] for i in sorted(completed)]
        metrics = compute_metrics(results)
        payload = {"metrics": metrics, "results": results,
                   "config": cfg_dump}
        config.output.results_file.parent.mkdir(parents=True, exist_ok=True)
        config.output.results_file.write_text(json.dumps(payload, indent=2))
        print(json.dumps(metrics, indent=2))
        return payload
    finally:
        if server is not None:
            server.stop()


def compute_metrics(results: list[dict]) -> dict:
    def acc(rows):
        if not rows:
            return None
        return sum(bool(r["grade"].get("correct")) for r in rows) / len(rows)

    mc = [r for r in results if r.get("type") == "mc"]
    qa = [r for r in results if r.get("type") == "qa"]
    rag_rows = [r for r in results if r["retrieval"].get("used_rag")]
    src_known = [r for r in rag_rows
                 if r["retrieval"].get("source_chunk_retrieved") is not None]
    return {
        "n": len(results),
        "overall_accuracy": acc(results),
        "mc_accuracy": acc(mc),
        "qa_accuracy": acc(qa),
        "grader_failures": sum(
            bool(r["grade"].get("grader_failed")) for r in results),
        "source_retrieval_rate": (
            sum(bool(r["retrieval"]["source_chunk_retrieved"])
                for r in src_known) / len(src_known)
            if src_known else None
        ),
        "question_hash_retrieval_rate": (
            sum(bool(r["retrieval"].get("question_hash_retrieved"))
                for r in rag_rows) / len(rag_rows)
            if rag_rows else None
        ),
        "mean_elapsed_s": (
            sum(r["elapsed_s"] for r in results) / len(results)
            if results else None
        ),
    }


def main() -> None:
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--config", type=Path, required=True)
    args = ap.parse_args()
    run_mcqa(MCQAConfig.from_yaml(args.config))


if __name__ == "__main__":
    main()
