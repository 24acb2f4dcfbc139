"""CPU tests for L6 applications: chat, OpenAI-compatible server,
MCQA harness components, CLI verbs."""

import json

import pytest
import yaml
from typer.testing import CliRunner

from distllm_amd.mcqa.harness import (
    Checkpointer,
    compute_metrics,
    find_free_port,
    make_chunk_id,
    parse_grade_json,
    question_hash,
)


def test_parse_grade_json_ladder():
    assert parse_grade_json('{"correct": true, "reason": "ok"}')["correct"]
    assert parse_grade_json(
        'Sure! Here: {"correct": false, "reason": "wrong"} hope that helps'
    )["correct"] is False
    assert parse_grade_json('I think "correct": true overall')["correct"]
    assert parse_grade_json("no json at all") is None


def test_chunk_id_and_hash_stability():
    a = make_chunk_id("/data/p1.jsonl", 7)
    assert a == make_chunk_id("/data/p1.jsonl", 7)
    assert a != make_chunk_id("/data/p2.jsonl", 7)
    assert len(a.split("_")[0]) == 16
    assert question_hash("Q?") == question_hash("Q?")


def test_checkpointer_roundtrip(tmp_path):
    ck = Checkpointer(tmp_path, tmp_path / "qs.json", "modelA")
    cfg = {"questions_file": str(tmp_path / "qs.json")}
    res = {0: {"answer": "a"}, 2: {"answer": "b"}}
    p = ck.save(res, cfg)
    assert p.exists()
    found = ck.find_latest()
    assert found == p
    loaded = ck.load(found, cfg)
    assert loaded == res
    # mismatched questions file -> ignored
    assert ck.load(found, {"questions_file": "other.json"}) == {}


def test_metrics():
    rows = [
        {"grade": {"correct": True}, "type": "mc",
         "retrieval": {"used_rag": True, "source_chunk_retrieved": True,
                       "question_hash_retrieved": False}, "elapsed_s": 1.0},
        {"grade": {"correct": False}, "type": "qa",
         "retrieval": {"used_rag": True, "source_chunk_retrieved": False,
                       "question_hash_retrieved": True}, "elapsed_s": 2.0},
    ]
    m = compute_metrics(rows)
    assert m["overall_accuracy"] == 0.5
    assert m["mc_accuracy"] == 1.0
    assert m["qa_accuracy"] == 0.0
    assert m["source_retrieval_rate"] == 0.5
    assert m["question_hash_retrieval_rate"] == 0.5


def test_find_free_port():
    p = find_free_port()
    assert 1024 <= p <= 65535


@pytest.fixture()
def chat_config(tmp_path):
    cfg = {
        "generator_config": {
            "name": "native", "model": "llama-tiny",
            "temperature": 0.0, "min_p": 0.0, "max_tokens": 4,
            "max_model_len": 256, "max_batch_size": 2,
        },
        "system_prompt": "Test assistant.",
        "transcript_dir": str(tmp_path / "transcripts"),
    }
    p = tmp_path / "chat.yaml"
    p.write_text(yaml.safe_dump(cfg))
    return p


def test_chat_session_and_transcript(chat_config, tmp_path):
    from distllm_amd.chat import ChatAppConfig, ChatSession

    session = ChatSession(ChatAppConfig.from_yaml(chat_config))
    ans = session.turn("Hello there")
    assert isinstance(ans, str)
    assert session.template.history[-1][0] == "assistant"
    out = session.save_transcript()
    assert out is not None and out.exists()


def test_chat_server_endpoint(chat_config, monkeypatch):
    from fastapi.testclient import TestClient

    import distllm_amd.chat_server as srv

    monkeypatch.setenv("DISTLLM_CHAT_CONFIG", str(chat_config))
    srv._session = None
    client = TestClient(srv.app)
    assert client.get("/health").json()["status"] == "ok"
    r = client.post("/v1/chat/completions", json={
        "messages": [{"role": "user", "content": "Hi"}],
    })
    assert r.status_code == 200
    body = r.json()
    assert body["choices"][0]["message"]["role"] == "assistant"
    # streaming shim
    r2 = client.post("/v1/chat/completions", json={
        "messages": [{"role": "user", "content": "Hi"}], "stream": True,
    })
    assert r2.status_code == 200
    assert "data:" in r2.text and "[DONE]" in r2.text


def test_env_substitution(monkeypatch):
    from distllm_amd.chat import substitute_env

    monkeypatch.setenv("MY_KEY", "sekrit")
    obj = {"a": "${env:MY_KEY}", "b": ["${env:MY_KEY}", 3]}
    out = substitute_env(obj)
    assert out == {"a": "sekrit", "b": ["sekrit", 3]}


def test_cli_embed_and_merge(tmp_path):
    from distllm_amd.cli import app

    runner = CliRunner()
    data = tmp_path / "in"
    data.mkdir()
    with open(data / "a.jsonl", "w") as fh:
        for i in range(3):
            fh.write(json.dumps({"text": f"Sample text {i}.",
                                 "path": f"p{i}"}) + "\n")
    out = tmp_path / "out"
    r = runner.invoke(app, [
        "embed", "--data-path", str(data), "--output-path", str(out),
        "--pretrained-model-name-or-path", "bert-tiny",
        "--dataset-name", "jsonl", "--embedder-name", "full_sequence",
        "--no-half-precision",
    ])
    assert r.exit_code == 0, r.output
    shards = [p for p in out.iterdir() if p.is_dir()]
    assert len(shards) == 1
    merged = tmp_path / "merged"
    r2 = runner.invoke(app, [
        "merge", "--dataset-dir", str(out), "--output-dir", str(merged),
    ])
    assert r2.exit_code == 0, r2.output
    from datasets import load_from_disk

    assert len(load_from_disk(str(merged))) == 3


def test_cli_chunk_fasta(tmp_path):
    from distllm_amd.cli import app

    runner = CliRunner()
    f = tmp_path / "x.fasta"
    f.write_text(">a\nMKV\n>b\nACD\n>c\nGLY\n>d\nWYQ\n")
    out = tmp_path / "chunks"
    r = runner.invoke(app, [
        "chunk-fasta-file", "--input-file", str(f), "--output-dir",
        str(out), "--num-chunks", "2",
    ])
    assert r.exit_code == 0, r.output
    assert len(list(out.glob("*.fasta"))) == 2


def test_prompt_postprocess_behaviors():
    from distllm_amd.generate import get_prompt_template

    qc = get_prompt_template({"name": "question_chunk"})
    out = qc.postprocess([
        "Here is context. What drives protein folding? It is known.",
        "No questions at all here.",
    ])
    assert out[0].strip().endswith("?")
    assert "folding" in out[0]
    assert out[1] == ""

    qa = get_prompt_template({"name": "question_answer"})
    assert qa.postprocess(["2. Mitochondria"]) == ["mitochondria"]
    assert qa.postprocess(["Option 3: DNA"]) == ["dna"]
    prompts = qa.preprocess(["Q?"], [["ctx a", "ctx b"]], [[0.9, 0.5]])
    assert "score: 0.9" in prompts[0] and "Q?" in prompts[0]

    ident = get_prompt_template({"name": "identity"})
    assert ident.preprocess("solo") == ["solo"]

    ks = get_prompt_template({"name": "keyword_selection",
                              "keywords": ["radiation", "dosimetry"]})
    p = ks.preprocess(["A paragraph."])[0]
    assert "radiation" in p and "A paragraph." in p

    amp = get_prompt_template({"name": "amp_question"})
    p2 = amp.preprocess(['{"name": "pep1", "function": "antimicrobial"}'])
    assert "pep1" in p2[0]
    import json as _json

    post = amp.postprocess(["Q: which peptide?"])
    assert _json.loads(post[0])["question"] == "Q: which peptide?"


def test_generate_worker_skips_empty_results(tmp_path, monkeypatch):
    """All responses dropped by postprocess -> no unloadable 0-row
    shard is written."""
    from datasets import Dataset as HFDataset

    from distllm_amd.distributed_generation import generate_worker

    HFDataset.from_dict({"text": ["short chunk"],
                         "path": ["p0"]}).save_to_disk(str(tmp_path / "in"))
    out = tmp_path / "gen"
    out.mkdir()
    # question_chunk postprocess drops answers without parseable output;
    # llama-tiny random weights produce none at max_tokens=2
    generate_worker(tmp_path / "in", out, {"name": "question_chunk"},
                    {"name": "huggingface"}, {"name": "huggingface"},
                    {"name": "native", "model": "llama-tiny",
                     "max_tokens": 2, "max_batch_size": 2})
    assert list(out.iterdir()) == []


def test_chat_target_dispatch(tmp_path, monkeypatch):
    """Reference chat_argoproxy `_target_` YAMLs load drop-in
    (chat_argoproxy.py:510-549 class-dispatch surface)."""
    import yaml

    from distllm_amd.chat import ChatAppConfig, translate_target_config

    monkeypatch.setenv("ARGO_KEY", "sekret")
    cfg = {
        "generator_config": {
            "_target_": "ArgoGeneratorConfig",
            "model": "argo:gpt-4o",
            "base_url": "http://localhost:56267",
            "api_key": "${env:ARGO_KEY}",
            "temperature": 0.0,
            "max_tokens": 128,
        },
        "system_prompt": "hi",
    }
    p = tmp_path / "chat.yaml"
    p.write_text(yaml.safe_dump(cfg))
    app = ChatAppConfig.from_yaml(p)
    g = app.generator_config
    assert g["name"] == "openai_http"
    assert g["base_url"] == "http://localhost:56267"
    assert g["api_key"] == "sekret"  # ${env:VAR} resolved
    assert g["model"] == "argo:gpt-4o" and g["max_tokens"] == 128

    v = translate_target_config(
        {"_target_": "VLLMGeneratorConfig", "server": "rbdgx1",
         "port": 8000, "api_key": "CELS", "model": "m"})
    assert v["base_url"] == "http://rbdgx1:8000/v1"

    o = translate_target_config(
        {"_target_": "OpenAIAPIGeneratorConfig", "model": "gpt-4.1",
         "api_key": "k"})
    assert o["base_url"] == "https://api.openai.com/v1"

    import pytest as _p
    with _p.raises(ValueError, match="Unknown generator config"):
        translate_target_config({"_target_": "NopeConfig"})


def test_mcqa_v2_yaml_translation(tmp_path):
    """Reference MCQA v2/v3-shaped YAML (nested generator_settings,
    rag.enabled, processing/output blocks, model_servers.yaml grader
    lookup) loads drop-in through MCQAConfig.from_yaml."""
    import yaml

    from distllm_amd.mcqa.harness import MCQAConfig

    (tmp_path / "model_servers.yaml").write_text(yaml.safe_dump({
        "servers": [{"shortname": "gpt-4.1",
                     "openai_api_base": "http://g:1/v1",
                     "openai_api_key": "k2",
                     "openai_model": "gpt-4.1-mini"}]}))
    cfg_path = tmp_path / "mcqa_v2.yaml"
    cfg_path.write_text(yaml.safe_dump({
        "questions_file": "/q.json",
        "model": {
            "generator": {"generator_type": "vllm"},
            "generator_settings": {
                "server": "rbdgx2", "model": "meta-llama/L70",
                "port": 9999, "api_key": "CELS",
                "temperature": 0.0, "max_tokens": 2048},
            "grader_shortname": "gpt-4.1",
            "model_config_file": "model_servers.yaml"},
        "rag": {"enabled": True, "rag_config_file": None,
                "retriever_config": None, "use_context_field": False,
                "retrieval_top_k": 5,
                "retrieval_score_threshold": 0.0,
                "chunk_logging_enabled": True},
        "processing": {"parallel_workers": 4, "question_format": "mc",
                       "verbose": True, "random_selection": 100,
                       "random_seed": None},
        "output": {"save_incorrect": True, "output_directory": "/tmp/o",
                   "output_prefix": "run1"},
    }))
    cfg = MCQAConfig.from_yaml(cfg_path)
    assert cfg.model.base_url == "http://rbdgx2:9999/v1"
    assert cfg.model.api_key == "CELS"
    assert cfg.model.max_tokens == 2048
    assert cfg.grader.base_url == "http://g:1/v1"
    assert cfg.grader.model == "gpt-4.1-mini"
    assert cfg.grader.api_key == "k2"
    assert cfg.rag.use_rag and cfg.rag.top_k == 5
    assert cfg.processing.random_selection == 100
    assert str(cfg.output.results_file) == "/tmp/o/run1_results.json"

    # native-shaped YAML passes through unchanged
    native = tmp_path / "native.yaml"
    native.write_text(yaml.safe_dump({
        "questions_file": "/q.json",
        "model": {"base_url": "http://x/v1"},
        "grader": {"base_url": "http://y/v1"}}))
    n = MCQAConfig.from_yaml(native)
    assert n.model.base_url == "http://x/v1"


def test_chat_reference_yaml_dialect(tmp_path):
    """Reference examples/chat/*.yaml shape (nested rag_configs block,
    bare server/port generator settings, save_conversation_path) loads
    drop-in through ChatAppConfig.from_yaml."""
    import yaml

    from distllm_amd.chat import ChatAppConfig

    p = tmp_path / "chat.yaml"
    p.write_text(yaml.safe_dump({
        "rag_configs": {
            "generator_config": {
                "server": "rbdgx2", "port": 9999,
                "model": "meta-llama/L70", "api_key": "CELS"},
            "retriever_config": None,
            "verbose": False},
        "save_conversation_path": str(tmp_path / "conv"),
    }))
    cfg = ChatAppConfig.from_yaml(p)
    g = cfg.generator_config
    assert g["name"] == "openai_http"
    assert g["base_url"] == "http://rbdgx2:9999/v1"
    assert g["api_key"] == "CELS" and g["model"] == "meta-llama/L70"
    assert str(cfg.transcript_dir) == str(tmp_path / "conv")


def test_cli_reference_flag_spellings(tmp_path):
    """The reference CLI's snake_case flags and short aliases
    (reference cli.py: --encoder_name/-mn, --chunk_size/-c, ...) work
    alongside the dash-case spellings."""
    from typer.testing import CliRunner

    from distllm_amd.cli import app

    fa = tmp_path / "x.fasta"
    fa.write_text(">a\nMKV\n>b\nMLL\n>c\nMAA\n>d\nMCC\n")
    out = tmp_path / "chunks"
    r = CliRunner().invoke(app, [
        "chunk-fasta-file", "--input_file", str(fa),
        "--output_dir", str(out), "--chunk_size", "2"])
    assert r.exit_code == 0, r.output
    assert len(list(out.glob("*.fasta"))) == 2
    out2 = tmp_path / "chunks2"
    r2 = CliRunner().invoke(app, [
        "chunk-fasta-file", "-i", str(fa), "-o", str(out2), "-c", "4"])
    assert r2.exit_code == 0, r2.output
