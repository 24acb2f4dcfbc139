"""End-to-end MCQA harness test: boots the real OpenAI-compatible
server subprocess (native llama-tiny engine), runs the 3-question
fixture through answering + grading + checkpointing."""

import json
from pathlib import Path

import pytest
import yaml

from distllm_amd.mcqa.harness import MCQAConfig, run_mcqa

pytestmark = pytest.mark.timeout(300)


def test_mcqa_end_to_end(tmp_path):
    chat_cfg = {
        "generator_config": {
            "name": "native", "model": "llama-tiny",
            "temperature": 0.0, "min_p": 0.0, "max_tokens": 8,
            "max_model_len": 512, "max_batch_size": 2,
        },
    }
    chat_path = tmp_path / "chat.yaml"
    chat_path.write_text(yaml.safe_dump(chat_cfg))

    cfg = MCQAConfig(
        questions_file=Path("tests/fixtures/test_questions_simple.json"),
        model={"boot_local": True, "local_config_path": str(chat_path),
               "temperature": 0.0, "max_tokens": 8},
        grader={"temperature": 0.0, "max_tokens": 8},
        processing={"parallel_workers": 2, "checkpoint_interval": 1},
        output={"results_file": str(tmp_path / "res.json"),
                "checkpoint_dir": str(tmp_path / "ckpt")},
    )
    payload = run_mcqa(cfg)
    assert payload["metrics"]["n"] == 3
    # grader is a random-weight model: accuracy is arbitrary but the
    # plumbing must produce a grade for every question
    assert all("grade" in r for r in payload["results"])
    assert (tmp_path / "res.json").exists()
    ckpts = list((tmp_path / "ckpt").glob("ckpt_*.json"))
    assert ckpts, "checkpoints written"

    # resume path: re-running skips completed questions
    payload2 = run_mcqa(cfg)
    assert payload2["metrics"]["n"] == 3


def test_mcqa_results_schema(tmp_path):
    data = json.loads(
        Path("tests/fixtures/test_questions_simple.json").read_text()
    )
    assert len(data) == 3
    assert all("question" in q and "answer" in q for q in data)
