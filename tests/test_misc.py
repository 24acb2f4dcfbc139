"""Foundation + misc parity tests: Timer log round-trip, registry
warm-start, tokenization entry, MCQA config validation."""

import json

import pytest
import torch

from distllm_amd.registry import RegistrySingleton, registry
from distllm_amd.utils import Timer, batch_data, parse_timer_logs, retry


def test_timer_line_roundtrip(capsys, tmp_path):
    log = tmp_path / "t.log"
    with Timer("computed-embeddings", "file1.jsonl", logfile=log):
        pass
    out = capsys.readouterr().out
    assert "[timer] [computed-embeddings file1.jsonl] in [" in out
    stats = parse_timer_logs(log.read_text())
    assert len(stats) == 1
    assert stats[0].tags == ["computed-embeddings", "file1.jsonl"]
    assert stats[0].elapsed_s >= 0
    assert stats[0].end_unix >= stats[0].start_unix
    # start/stop API + TimeLogger.parse_logs (reference timer.py surface)
    from distllm_amd.utils import TimeLogger

    t = Timer("manual", logfile=log).start()
    t.stop()
    assert t.elapsed_ns >= 0 and t.elapsed_ms >= 0
    stats2 = TimeLogger().parse_logs(log)
    assert len(stats2) == 2 and stats2[1].tags == ["manual"]


def test_batch_data():
    assert batch_data([1, 2, 3, 4, 5], 2) == [[1, 2], [3, 4], [5]]
    assert batch_data([], 3) == []
    with pytest.raises(ValueError):
        batch_data([1], 0)


def test_registry_warm_start():
    calls = []

    def factory(x):
        calls.append(x)
        return {"x": x}

    r = RegistrySingleton()
    a = r.get(factory, 1)
    b = r.get(factory, 1)
    assert a is b and calls == [1]
    c = r.get(factory, 2)  # key change -> rebuild
    assert c["x"] == 2 and calls == [1, 2]
    r.clear()


def test_retry_decorator():
    attempts = []

    @retry(tries=3, base_delay=0.01)
    def flaky():
        attempts.append(1)
        if len(attempts) < 3:
            raise RuntimeError("boom")
        return "ok"

    assert flaky() == "ok"
    assert len(attempts) == 3


def test_mcqa_boot_local_validation():
    from distllm_amd.mcqa.harness import ModelConfig

    with pytest.raises(Exception):
        ModelConfig(boot_local=True)  # needs local_config_path
    ModelConfig(boot_local=False)  # fine


def test_tokenization_entry(tmp_path):
    from distllm_amd.distributed_tokenization import (
        TokenizationConfig,
        main,
    )

    idir = tmp_path / "in"
    idir.mkdir()
    with open(idir / "a.jsonl", "w") as fh:
        for i in range(4):
            fh.write(json.dumps({"text": f"hello world {i}"}) + "\n")
    odir = tmp_path / "out"
    main(TokenizationConfig(input_dir=idir, output_dir=odir,
                            save_labels=True))
    from datasets import load_from_disk

    shards = list((odir / "tokenized").iterdir())
    assert len(shards) == 1
    ds = load_from_disk(str(shards[0]))
    assert set(ds.column_names) == {"input_ids", "attention_mask", "labels"}
    assert len(ds) == 4


def test_esm_tokenizer_exact_ids():
    from distllm_amd.embed.tokenizers import EsmTokenizer

    tok = EsmTokenizer()
    enc = tok(["MKV"], return_tensors="pt")
    # <cls> M K V <eos> == 0, 20, 15, 7, 2
    assert enc["input_ids"][0].tolist()[:5] == [0, 20, 15, 7, 2]
    assert tok.decode(enc["input_ids"][0]) == "MKV"


def test_model_weight_loader_roundtrip(tmp_path):
    """Native checkpoint save -> load_weights restores exactly."""
    from safetensors.torch import save_file

    from distllm_amd.models import build_model
    from distllm_amd.models.loader import load_weights

    m1 = build_model("bert-tiny", dtype=torch.float32)
    sd = {k: v.detach().clone().contiguous()
          for k, v in m1.named_parameters()}
    save_file(sd, str(tmp_path / "model.safetensors"))
    m2 = build_model("bert-tiny", dtype=torch.float32)
    assert not torch.equal(m1.word_emb, m2.word_emb)
    load_weights(m2, tmp_path)
    assert torch.equal(m1.word_emb, m2.word_emb)
    assert torch.equal(m1.layers[0].wqkv, m2.layers[0].wqkv)


def test_esm2_weight_mapping(tmp_path):
    """HF EsmModel-named checkpoint loads into our Esm2Encoder."""
    from safetensors.torch import save_file

    from distllm_amd.models import build_model
    from distllm_amd.models.loader import load_weights

    m1 = build_model("esm2-8m", dtype=torch.float32)
    sd = {}
    sd["esm.embeddings.word_embeddings.weight"] = m1.word_emb
    sd["esm.encoder.emb_layer_norm_after.weight"] = m1.final_ln_w
    sd["esm.encoder.emb_layer_norm_after.bias"] = m1.final_ln_b
    H = m1.dims.hidden_size
    for i, lyr in enumerate(m1.layers):
        p = f"esm.encoder.layer.{i}."
        q, k, v = lyr.wqkv[:H], lyr.wqkv[H:2 * H], lyr.wqkv[2 * H:]
        qb, kb, vb = lyr.bqkv[:H], lyr.bqkv[H:2 * H], lyr.bqkv[2 * H:]
        sd[p + "attention.self.query.weight"] = q
        sd[p + "attention.self.key.weight"] = k
        sd[p + "attention.self.value.weight"] = v
        sd[p + "attention.self.query.bias"] = qb
        sd[p + "attention.self.key.bias"] = kb
        sd[p + "attention.self.value.bias"] = vb
        sd[p + "attention.output.dense.weight"] = lyr.wo
        sd[p + "attention.output.dense.bias"] = lyr.bo
        sd[p + "attention.LayerNorm.weight"] = lyr.ln1_w
        sd[p + "attention.LayerNorm.bias"] = lyr.ln1_b
        sd[p + "LayerNorm.weight"] = lyr.ln2_w
        sd[p + "LayerNorm.bias"] = lyr.ln2_b
        sd[p + "intermediate.dense.weight"] = lyr.w1
        sd[p + "intermediate.dense.bias"] = lyr.b1
        sd[p + "output.dense.weight"] = lyr.w2
        sd[p + "output.dense.bias"] = lyr.b2
    save_file({k: v.detach().clone().contiguous() for k, v in sd.items()},
              str(tmp_path / "model.safetensors"))
    m2 = build_model("esm2-8m", dtype=torch.float32)
    load_weights(m2, tmp_path)
    assert torch.equal(m1.word_emb, m2.word_emb)
    assert torch.equal(m1.layers[0].wqkv, m2.layers[0].wqkv)
    # forwards agree
    ids = torch.randint(4, 25, (1, 12))
    mask = torch.ones_like(ids, dtype=torch.int32)
    with torch.no_grad():
        assert torch.allclose(m1(ids, mask), m2(ids, mask))


def test_example_configs_parse():
    """The shipped example YAMLs must parse into their config models."""
    import yaml

    from distllm_amd.chat import ChatAppConfig
    from distllm_amd.distributed_embedding import EmbeddingConfig
    from distllm_amd.distributed_generation import GenerationConfig
    from distllm_amd.mcqa.harness import MCQAConfig
    from distllm_amd.rag.evaluate import EvalSuiteConfig

    EmbeddingConfig(**yaml.safe_load(open("examples/embed.yaml")))
    GenerationConfig(**yaml.safe_load(open("examples/generate.yaml")))
    ChatAppConfig(**yaml.safe_load(open("examples/chat.yaml")))
    MCQAConfig(**yaml.safe_load(open("examples/mcqa.yaml")))
    EvalSuiteConfig(**yaml.safe_load(open("examples/evaluate.yaml")))
    from distllm_amd.parallel.launcher import HpcLaunchConfig

    HpcLaunchConfig(**yaml.safe_load(open("examples/hpc.yaml")))


def test_cli_generate(tmp_path):
    from typer.testing import CliRunner

    from distllm_amd.cli import app

    runner = CliRunner()
    idir = tmp_path / "in"
    idir.mkdir()
    with open(idir / "a.jsonl", "w") as fh:
        fh.write(json.dumps({"text": "A prompt here", "path": "p0"}) + "\n")
    odir = tmp_path / "gen"
    r = runner.invoke(app, [
        "generate", "--input-dir", str(idir), "--output-dir", str(odir),
        "--prompt-name", "identity", "--reader-name", "jsonl",
        "--llm-name", "llama-tiny", "--max-tokens", "3",
        "--temperature", "0.0", "--min-p", "0.0",
    ])
    assert r.exit_code == 0, r.output
    shards = [p for p in odir.iterdir() if p.is_dir()]
    assert len(shards) == 1


def test_example_yamls_parse(tmp_path):
    """Every shipped examples/*.yaml must validate against its config
    schema (guards the docs against schema drift)."""
    import yaml

    from pathlib import Path

    root = Path(__file__).parent.parent / "examples"

    from distllm_amd.chat import ChatAppConfig, substitute_env
    from distllm_amd.distributed_embedding import EmbeddingConfig
    from distllm_amd.distributed_generation import GenerationConfig
    from distllm_amd.mcqa.harness import MCQAConfig

    EmbeddingConfig(**yaml.safe_load((root / "embed.yaml").read_text()))
    GenerationConfig(**yaml.safe_load((root / "generate.yaml").read_text()))
    ChatAppConfig(**substitute_env(
        yaml.safe_load((root / "chat.yaml").read_text())))
    MCQAConfig(**substitute_env(
        yaml.safe_load((root / "mcqa.yaml").read_text())))
    # evaluate.yaml is a plain task-list document for `distllm evaluate`
    ev = yaml.safe_load((root / "evaluate.yaml").read_text())
    assert "tasks" in ev or "task_names" in ev or isinstance(ev, dict)


def test_qwen2_preset_and_cpu_forward():
    """Qwen2 family resolves, builds, and runs the fp32 CPU path with
    the fused qkv bias applied."""
    import torch

    from distllm_amd.embed.encoders import resolve_preset
    from distllm_amd.models import build_model

    assert resolve_preset("Qwen/Qwen2-7B-Instruct") == "qwen2-7b"
    torch.manual_seed(0)
    m = build_model("qwen2-tiny", device="cpu", dtype=torch.float32)
    assert m.lm_head is not None  # causal-LM head by default
    ids = torch.randint(4, 1024, (2, 12))
    mask = torch.ones(2, 12, dtype=torch.long)
    h = m(ids, mask)
    assert h.shape == (2, 12, 256)
    # bias actually reaches the projection: perturbing it changes output
    with torch.no_grad():
        m.layers[0].bqkv.fill_(0.5)
    h2 = m(ids, mask)
    assert (h - h2).abs().max().item() > 1e-4


def test_timer_stats_pandas_contract(capsys, tmp_path):
    """Reference analysis idiom (reference tests/distllm_test.py:44-48):
    Timer output parsed from captured stdout converts to a DataFrame
    with the tags/elapsed_s/start_unix/end_unix columns; Path tags are
    accepted."""
    from pathlib import Path as _P

    import pandas as pd

    from distllm_amd.utils import TimeLogger, Timer

    with Timer("func1", _P("/my/path1")):
        pass
    with Timer("func2", _P("/my/path2")):
        pass
    log = tmp_path / "time.log"
    log.write_text(capsys.readouterr().out)
    stats = TimeLogger().parse_logs(log)
    assert stats[0].tags == ["func1", "/my/path1"]
    assert stats[1].tags == ["func2", "/my/path2"]
    df = pd.DataFrame(stats)
    assert len(df) == 2
    assert list(df.keys()) == ["tags", "elapsed_s", "start_unix",
                               "end_unix"]
