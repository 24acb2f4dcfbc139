"""Full user workflow ON GPU through the real CLI: embed a jsonl corpus
with the HIP encoder, merge the shards, then run the native generate
pass over the merged dataset — the reference's primary pipeline
(distributed_embedding -> merge -> distributed_generation) end-to-end
via `distllm` commands (reference: distllm/cli.py + SURVEY.md §3.1/3.2).
Small models (bert-tiny / llama-tiny, random weights) keep it fast; the
point is that the CLI surface drives the HIP compute path."""

import json

import pytest
from typer.testing import CliRunner

pytestmark = pytest.mark.gpu


def test_cli_embed_merge_generate_gpu(tmp_path):
    from datasets import load_from_disk

    from distllm_amd.cli import app

    runner = CliRunner()
    data = tmp_path / "in"
    data.mkdir()
    # sentences long enough to survive jsonl_chunk's 750-char
    # min_buffer_length (reference default) after 3-sentence windows
    def sent(i, j):
        body = " ".join(
            f"observation {i}-{j}-{k} of the binding assay shows a "
            f"measurable affinity shift across the mutant panel"
            for k in range(4)
        )
        return f"Study {i} section {j} reports that {body}."

    with open(data / "docs.jsonl", "w") as fh:
        for i in range(4):
            fh.write(json.dumps({
                "text": " ".join(sent(i, j) for j in range(6)),
                "path": f"doc{i}",
            }) + "\n")

    out = tmp_path / "out"
    r = runner.invoke(app, [
        "embed", "--data-path", str(data), "--output-path", str(out),
        "--pretrained-model-name-or-path", "bert-tiny",
        "--dataset-name", "jsonl_chunk", "--embedder-name",
        "semantic_chunk", "--batch-size", "8",
    ])
    assert r.exit_code == 0, r.output

    merged = tmp_path / "merged"
    r2 = runner.invoke(app, [
        "merge", "--dataset-dir", str(out), "--output-dir", str(merged),
    ])
    assert r2.exit_code == 0, r2.output
    ds = load_from_disk(str(merged))
    assert len(ds) >= 1
    assert "embeddings" in ds.column_names

    gen = tmp_path / "gen"
    r3 = runner.invoke(app, [
        "generate", "--input-dir", str(merged), "--output-dir", str(gen),
        "--llm-name", "llama-tiny", "--generator-name", "native",
        "--prompt-name", "identity", "--max-tokens", "8",
        "--batch-size", "4",
    ])
    assert r3.exit_code == 0, r3.output
    shards = [p for p in gen.iterdir() if p.is_dir()]
    assert len(shards) == 1
    gds = load_from_disk(str(shards[0]))
    assert "response" in gds.column_names
    assert len(gds) == len(ds)
