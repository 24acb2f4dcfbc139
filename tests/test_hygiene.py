"""Defaults-hygiene and ADVICE-fix regression tests (round 2).

Covers: HashWordTokenizer opt-in gating, ubinary recall warning,
huggingface-alias decoding defaults, IVF -1 index filtering, beam
length clamping, seeded top-p sampling determinism.
"""

from __future__ import annotations

import warnings

import pytest
import torch

from distllm_amd.embed.tokenizers import HashWordTokenizer, load_tokenizer


def test_hash_fallback_requires_opt_in(monkeypatch):
    monkeypatch.delenv("DISTLLM_ALLOW_HASH_TOKENIZER", raising=False)
    with pytest.raises(ValueError, match="HashWordTokenizer"):
        load_tokenizer("no-such-model-anywhere", arch="bert")


def test_hash_fallback_warns_when_opted_in(monkeypatch):
    monkeypatch.delenv("DISTLLM_ALLOW_HASH_TOKENIZER", raising=False)
    with pytest.warns(UserWarning, match="HashWordTokenizer"):
        tok = load_tokenizer("no-such-model-anywhere", arch="bert",
                             allow_hash_fallback=True)
    assert isinstance(tok, HashWordTokenizer)


def test_hash_fallback_env_opt_in(monkeypatch):
    monkeypatch.setenv("DISTLLM_ALLOW_HASH_TOKENIZER", "1")
    with pytest.warns(UserWarning):
        tok = load_tokenizer("no-such-model-anywhere", arch="bert")
    assert isinstance(tok, HashWordTokenizer)


def test_esm_arch_never_uses_hash(monkeypatch):
    monkeypatch.delenv("DISTLLM_ALLOW_HASH_TOKENIZER", raising=False)
    tok = load_tokenizer("esm2-any", arch="esm2")  # no raise, no warn
    assert not isinstance(tok, HashWordTokenizer)


def test_ubinary_warns(tmp_path):
    import numpy as np
    from datasets import Dataset

    from distllm_amd.rag.search import HipFlatIndex

    rng = np.random.default_rng(0)
    emb = rng.standard_normal((64, 32)).astype(np.float32)
    ds = Dataset.from_dict(
        {"text": [f"t{i}" for i in range(64)], "embeddings": emb.tolist()}
    )
    dsd = tmp_path / "ds"
    ds.save_to_disk(str(dsd))
    with pytest.warns(UserWarning, match="ubinary"):
        idx = HipFlatIndex(dsd, tmp_path / "idx.npz", precision="ubinary",
                           device="cpu")
    assert idx.rescore_multiplier == 8  # raised default


def test_huggingface_alias_defaults():
    from distllm_amd.generate.generators import NativeGeneratorConfig

    hf = NativeGeneratorConfig(name="huggingface")
    assert hf.num_beams == 10 and hf.top_p == 0.95
    # explicit values are never overridden
    hf2 = NativeGeneratorConfig(name="huggingface", num_beams=1, top_p=0.5)
    assert hf2.num_beams == 1 and hf2.top_p == 0.5
    nat = NativeGeneratorConfig(name="native")
    assert nat.num_beams == 0


def test_search_filters_padded_indices(tmp_path):
    """IVF pads short candidate lists with -1; search() must drop them
    even when score_threshold is 0 (the default)."""
    import numpy as np
    from datasets import Dataset

    from distllm_amd.rag.search import BatchedSearchResults, HipFlatIndex

    rng = np.random.default_rng(1)
    emb = rng.standard_normal((32, 16)).astype(np.float32)
    ds = Dataset.from_dict(
        {"text": [f"t{i}" for i in range(32)], "embeddings": emb.tolist()}
    )
    dsd = tmp_path / "ds"
    ds.save_to_disk(str(dsd))
    idx = HipFlatIndex(dsd, tmp_path / "idx.npz", device="cpu")

    # monkey in a scorer that returns a padded row
    def fake_topk(q, k):
        v = torch.full((q.shape[0], k), float("-inf"))
        i = torch.full((q.shape[0], k), -1, dtype=torch.int32)
        v[:, 0] = 1.0
        i[:, 0] = 3
        return v, i

    idx._topk_exact = fake_topk
    res = idx.search(rng.standard_normal((2, 16)).astype(np.float32),
                     top_k=4)
    assert isinstance(res, BatchedSearchResults)
    for row in res.total_indices:
        assert row == [3]
    for srow in res.total_scores:
        assert srow == [1.0]


def test_beam_clamps_to_max_positions():
    from distllm_amd.engine.beam import beam_generate
    from distllm_amd.models import build_model

    torch.manual_seed(0)
    model = build_model("llama-tiny", device="cpu", dtype=torch.float32)
    max_pos = model.dims.max_position_embeddings
    long_prompt = [5] * (max_pos + 40)  # longer than the model allows
    out = beam_generate(model, long_prompt, num_beams=2, max_tokens=8)
    assert len(out) <= 8  # did not crash on RoPE range


def test_top_p_sampling_seeded_and_min_p():
    """_sample_top_p must use the engine RNG (deterministic across
    identically-seeded engines, i.e. TP ranks) and honor min_p."""
    from distllm_amd.engine.engine import LLMEngine, SamplingParams

    class _E:  # minimal shim carrying only what _sample_top_p reads
        _rng = torch.Generator().manual_seed(0x5EED)

    params = SamplingParams(temperature=1.0, top_p=0.9, min_p=0.0)
    logits = torch.randn(4, 64, generator=torch.Generator().manual_seed(7))
    a = LLMEngine._sample_top_p(_E(), logits, params)
    _E._rng = torch.Generator().manual_seed(0x5EED)
    b = LLMEngine._sample_top_p(_E(), logits, params)
    assert torch.equal(a, b)  # same seed -> same tokens (rank parity)

    # min_p: with a spiked distribution and min_p close to 1, only the
    # argmax survives the floor regardless of top_p
    spiked = torch.full((8, 16), -10.0)
    spiked[:, 5] = 10.0
    params2 = SamplingParams(temperature=1.0, top_p=1.0 - 1e-9, min_p=0.99)
    _E._rng = torch.Generator().manual_seed(1)
    toks = LLMEngine._sample_top_p(_E(), spiked, params2)
    assert (toks == 5).all()


def test_quantization_flag_cpu_warns_and_stays_bf16():
    """Reference YAMLs with `quantization: true` (NF4 upstream) map to
    the fp8-e4m3 weight path on GPU; off-GPU the flag warns and the
    generator still works in bf16."""
    import warnings

    from distllm_amd.generate.generators import (
        NativeGenerator,
        NativeGeneratorConfig,
    )

    cfg = NativeGeneratorConfig(name="native", model="llama-tiny",
                                quantization=True, max_tokens=4,
                                max_model_len=128)
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        gen = NativeGenerator(cfg)
        assert any("quantization" in str(x.message) for x in w)
    assert gen.engine.weight_dtype == "bf16"
    out = gen.generate(["hello there"])
    assert len(out) == 1


def test_reference_generator_dialects_parse():
    """Reference vllm/huggingface backend YAML fields parse drop-in
    (vllm_backend.py llm_name; huggingface_backend.py
    pretrained_model_name_or_path/batch_size/do_sample + inert runtime
    knobs)."""
    from distllm_amd.generate.generators import NativeGeneratorConfig

    v = NativeGeneratorConfig(
        name="vllm", llm_name="mistralai/Mistral-7B-Instruct-v0.2",
        top_p=0.95, tensor_parallel_size=1, trust_remote_code=True)
    assert v.model == "mistralai/Mistral-7B-Instruct-v0.2"

    h = NativeGeneratorConfig(
        name="huggingface",
        pretrained_model_name_or_path="meta-llama/Llama-3-8b",
        half_precision=True, eval_mode=True, compile_model=False,
        quantization=False, do_sample=False, batch_size=4)
    assert h.model == "meta-llama/Llama-3-8b"
    assert h.max_batch_size == 4
    assert h.temperature == 0.0  # do_sample: false -> greedy
    assert h.num_beams == 10     # reference hf defaults still apply


def test_reference_encoder_dialects_parse():
    """Reference encoder YAML shapes (incl. the shipped examples'
    tokenizer_name spelling and esm eval_mode/compile_model) parse
    drop-in."""
    from distllm_amd.embed.encoders import (
        AutoEncoderConfig,
        Esm2EncoderConfig,
        EsmCambrianEncoderConfig,
    )

    a = AutoEncoderConfig(
        pretrained_model_name_or_path="some/checkpoint-63423",
        tokenizer_name="bert-base-uncased", quantization=False)
    assert a.tokenizer_path == "bert-base-uncased"
    e = Esm2EncoderConfig(eval_mode=True, compile_model=False,
                          faesm=True)
    assert e.half_precision
    EsmCambrianEncoderConfig(eval_mode=True, compile_model=False)


def test_faiss_v1_yaml_shape_parses():
    """Reference FaissIndexV1 YAML blocks (examples/evaluate/*.yaml)
    give only dataset_dir; the index path defaults alongside it."""
    from distllm_amd.rag.search import HipIndexConfig

    c = HipIndexConfig(name="faiss_index_v1", dataset_dir="/data/m")
    assert str(c.faiss_index_path) == "/data/m/hip_index.npz"
