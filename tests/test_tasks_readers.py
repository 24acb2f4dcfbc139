"""CPU tests for the eval-suite tasks (offline: data files pre-placed
so curl_download is skipped — reference distllm/rag/tasks/*) and the
generate readers."""

import json

import pytest

from distllm_amd.rag.tasks import ABSTAIN, TASKS, get_task


def _place(task, payload: str) -> None:
    task.data_file.write_text(payload)


def test_litqa_load_and_metrics(tmp_path):
    t = get_task("litqa", tmp_path)
    _place(t, "\n".join(json.dumps({
        "question": f"What binds target {i}",
        "ideal": f"Ligand{i}",
        "distractors": ["foo", "bar", "baz", "qux"],
    }) for i in range(4)))
    qs, ans = t.load_data()
    assert len(qs) == len(ans) == 4
    assert all(a.startswith("ligand") for a in ans)
    assert "Options:" in qs[0] and qs[0].count("\n") >= 5
    # metrics: accuracy counts abstains as wrong, precision excludes them
    preds = [ans[0], "wrong", ABSTAIN, ans[3]]
    assert t.compute_accuracy(ans, preds) == pytest.approx(0.5)
    assert t.compute_precision(ans, preds) == pytest.approx(2 / 3)


def test_sciq_load(tmp_path):
    t = get_task("sciq", tmp_path)
    _place(t, json.dumps([{
        "question": "What is water",
        "correct_answer": "H2O",
        "distractor1": "CO2", "distractor2": "O2", "distractor3": "N2",
    }]))
    qs, ans = t.load_data()
    assert ans == ["h2o"]
    assert qs[0].startswith("What is water?")


def test_pubmedqa_load(tmp_path):
    t = get_task("pubmedqa", tmp_path)
    _place(t, json.dumps({"id1": {
        "QUESTION": "Does X help",
        "CONTEXTS": ["ctx a", "ctx b"],
        "final_decision": "Yes",
    }}))
    qs, ans = t.load_data()
    assert ans == ["yes"]
    assert "Most relevant context:" in qs[0]
    assert "1. yes" in qs[0]


def test_protein_tasks_load(tmp_path):
    t = get_task("protein_function_qa", tmp_path)
    long_q = " ".join(["word"] * 300)
    _place(t, "\n".join([
        json.dumps({"question": "What does P1 do",
                    "options": ["a", "b"], "answer": "A"}),
        json.dumps({"question": long_q, "options": [], "answer": "x"}),
    ]))
    qs, ans = t.load_data()
    assert len(qs) == 1 and ans == ["a"]  # long question filtered

    t2 = get_task("protein_interaction_qa", tmp_path)
    _place(t2, json.dumps([{"question": "Does P1 bind P2",
                            "options": ["yes", "no"], "answer": "yes"}]))
    qs2, ans2 = t2.load_data()
    assert len(qs2) == 1 and ans2 == ["yes"]


def test_get_task_unknown(tmp_path):
    with pytest.raises(ValueError, match="unknown task"):
        get_task("nope", tmp_path)
    assert set(TASKS) == {"litqa", "sciq", "pubmedqa",
                          "protein_function_qa",
                          "protein_interaction_qa"}


def test_jsonl_reader(tmp_path):
    from distllm_amd.generate.readers import get_reader

    f = tmp_path / "x.jsonl"
    f.write_text("\n".join(json.dumps({"text": f"t{i}", "path": f"p{i}"})
                           for i in range(3)))
    rd = get_reader({"name": "jsonl"})
    text, paths = rd.read(f)
    assert text == ["t0", "t1", "t2"]
    assert len(paths) == 3


def test_huggingface_reader_roundtrip(tmp_path):
    from datasets import Dataset as HFDataset

    from distllm_amd.generate.readers import get_reader

    HFDataset.from_dict({
        "text": ["a", "b"], "path": ["p0", "p1"],
    }).save_to_disk(str(tmp_path / "ds"))
    rd = get_reader({"name": "huggingface"})
    text, paths = rd.read(tmp_path / "ds")
    assert text == ["a", "b"] and paths == ["p0", "p1"]


def test_api_generator_wire_formats(tmp_path, monkeypatch):
    """The `langchain`-parity API generator speaks the three wire
    protocols (OpenAI / Google Generative Language / Anthropic
    Messages) without SDKs; request construction is checked offline."""
    from distllm_amd.generate import get_generator

    envf = tmp_path / "env"
    envf.write_text("OPENAI_API_KEY=okey\nGOOGLE_API_KEY=gkey\n"
                    "ANTHROPIC_API_KEY=akey\n")
    monkeypatch.delenv("OPENAI_API_KEY", raising=False)
    monkeypatch.delenv("GOOGLE_API_KEY", raising=False)
    monkeypatch.delenv("ANTHROPIC_API_KEY", raising=False)

    g = get_generator({"name": "langchain", "llm": "gpt-3.5-turbo",
                       "dotenv_path": str(envf), "temperature": 0.0})
    url, hdr, body, parse = g.build_request("hi")
    assert url.endswith("/v1/chat/completions")
    assert hdr["Authorization"] == "Bearer okey"
    assert body["messages"] == [{"role": "user", "content": "hi"}]
    assert parse({"choices": [{"message": {"content": "x"}}]}) == "x"

    g2 = get_generator({"name": "langchain", "llm": "gemini-pro",
                        "dotenv_path": str(envf)})
    url, hdr, body, parse = g2.build_request("q")
    assert "generativelanguage.googleapis.com" in url and "key=gkey" in url
    assert body["contents"][0]["parts"][0]["text"] == "q"
    assert parse({"candidates": [{"content": {"parts": [{"text": "y"}]}}]}) == "y"

    g3 = get_generator({"name": "langchain",
                        "llm": "claude-3-opus-20240229",
                        "dotenv_path": str(envf)})
    url, hdr, body, parse = g3.build_request("p")
    assert url == "https://api.anthropic.com/v1/messages"
    assert hdr["x-api-key"] == "akey"
    assert hdr["anthropic-version"] == "2023-06-01"
    assert parse({"content": [{"text": "z"}]}) == "z"
