"""CPU end-to-end tests of the embed pipeline (reference call stack
SURVEY.md §3.1) on tiny fixtures — no GPU required."""

import json

import numpy as np
import pytest

from distllm_amd.distributed_embedding import EmbeddingConfig, main
from distllm_amd.embed import (
    get_dataset,
    get_embedder,
    get_encoder,
    get_pooler,
    get_writer,
)
from distllm_amd.embed.splitter import sentences_to_buffers, split_sentences


def _write_jsonl(path, n=6, words=200):
    with open(path, "w") as fh:
        for i in range(n):
            text = " ".join(
                f"W{i}x{j} word{j} is here and sentence {j} ends."
                for j in range(words // 8)
            )
            fh.write(json.dumps({"text": text, "path": f"doc{i}"}) + "\n")


def test_splitter_roundtrip():
    text = ("Dr. Smith went to Washington. He arrived at 5 p.m. on Tuesday! "
            "Was it raining? Yes. The experiment (Fig. 3) shows results.")
    sents = split_sentences(text)
    assert "".join(sents) == text
    assert len(sents) >= 4
    bufs = sentences_to_buffers(sents, 1)
    assert len(bufs) == len(sents)
    # middle buffer spans 3 sentences
    if len(sents) >= 3:
        assert sents[0] in bufs[1] and sents[2] in bufs[1]


def test_full_sequence_pipeline(tmp_path):
    f = tmp_path / "a.jsonl"
    _write_jsonl(f, n=5)
    enc = get_encoder({"name": "auto",
                       "pretrained_model_name_or_path": "bert-tiny",
                       "half_precision": False})
    ds = get_dataset({"name": "jsonl", "batch_size": 2,
                      "num_data_workers": 0, "pin_memory": False})
    pooler = get_pooler({"name": "mean"})
    embedder = get_embedder({"name": "full_sequence",
                             "normalize_embeddings": True})
    loader = ds.get_dataloader(f, enc)
    result = embedder.embed(loader, enc, pooler)
    assert result.embeddings.shape == (5, enc.embedding_size)
    norms = np.linalg.norm(result.embeddings, axis=1)
    assert np.allclose(norms, 1.0, atol=1e-3)
    assert result.metadata[0]["path"] == "doc0"


def test_semantic_chunk_pipeline(tmp_path):
    f = tmp_path / "a.jsonl"
    _write_jsonl(f, n=3, words=400)
    enc = get_encoder({"name": "auto",
                       "pretrained_model_name_or_path": "bert-tiny",
                       "half_precision": False})
    ds = get_dataset({"name": "jsonl_chunk", "batch_size": 4,
                      "num_data_workers": 0, "pin_memory": False,
                      "min_buffer_length": 20, "buffer_size": 1})
    pooler = get_pooler({"name": "mean"})
    embedder = get_embedder({"name": "semantic_chunk",
                             "chunk_batch_size": 4,
                             "min_chunk_length": 10,
                             "breakpoint_percentile_threshold": 80})
    loader = ds.get_dataloader(f, enc)
    result = embedder.embed(loader, enc, pooler)
    assert result.embeddings.shape[0] == len(result.text)
    assert result.embeddings.shape[0] >= 3  # at least one chunk per doc
    assert all("sentence" not in md for md in result.metadata)


def test_semantic_chunk_determinism(tmp_path):
    f = tmp_path / "a.jsonl"
    _write_jsonl(f, n=2, words=300)
    def run():
        enc = get_encoder({"name": "auto",
                           "pretrained_model_name_or_path": "bert-tiny",
                           "half_precision": False}, register=True)
        ds = get_dataset({"name": "jsonl_chunk", "batch_size": 4,
                          "num_data_workers": 0, "pin_memory": False,
                          "min_buffer_length": 20})
        pooler = get_pooler({"name": "mean"})
        embedder = get_embedder({"name": "semantic_chunk",
                                 "chunk_batch_size": 4,
                                 "min_chunk_length": 10})
        return embedder.embed(ds.get_dataloader(f, enc), enc, pooler)
    r1, r2 = run(), run()
    assert r1.text == r2.text  # breakpoints are deterministic
    assert np.allclose(r1.embeddings, r2.embeddings)


def test_fasta_pipeline(tmp_path):
    f = tmp_path / "p.fasta"
    f.write_text(">seq1 desc\nMKVLAT\nGLYQ\n>seq2\nACDEFGHIKLMNPQRSTVWY\n")
    enc = get_encoder({"name": "esm2",
                       "pretrained_model_name_or_path": "esm2-8m",
                       "half_precision": False})
    ds = get_dataset({"name": "fasta", "batch_size": 2,
                      "num_data_workers": 0, "pin_memory": False})
    pooler = get_pooler({"name": "mean"})
    embedder = get_embedder({"name": "full_sequence"})
    result = embedder.embed(ds.get_dataloader(f, enc), enc, pooler)
    assert result.embeddings.shape == (2, 320)
    assert result.metadata[0]["tags"] == "seq1 desc"


def test_writers_roundtrip(tmp_path):
    from distllm_amd.embed.embedders import EmbedderResult

    res = EmbedderResult(
        embeddings=np.random.rand(4, 8).astype(np.float32),
        text=[f"t{i}" for i in range(4)],
        metadata=[{"path": f"p{i}"} for i in range(4)],
    )
    # numpy
    w = get_writer({"name": "numpy"})
    w.write(tmp_path / "np1", res)
    w.write(tmp_path / "np2", res)
    w.merge([tmp_path / "np1", tmp_path / "np2", tmp_path / "missing"],
            tmp_path / "npm")
    m = np.load(tmp_path / "npm" / "embeddings.npy")
    assert m.shape == (8, 8)
    # huggingface (the interchange format: text, embeddings, metadata cols)
    w = get_writer({"name": "huggingface"})
    w.write(tmp_path / "hf1", res)
    w.write(tmp_path / "hf2", res)
    w.merge([tmp_path / "hf1", tmp_path / "hf2"], tmp_path / "hfm")
    from datasets import load_from_disk

    ds = load_from_disk(str(tmp_path / "hfm"))
    assert set(ds.column_names) == {"text", "embeddings", "path"}
    assert len(ds) == 8


def test_distributed_embedding_entry(tmp_path):
    idir = tmp_path / "in"
    odir = tmp_path / "out"
    idir.mkdir()
    for i in range(2):
        _write_jsonl(idir / f"f{i}.jsonl", n=3)
    cfg = EmbeddingConfig(
        input_dir=idir,
        output_dir=odir,
        glob_patterns=["*.jsonl"],
        dataset_config={"name": "jsonl", "batch_size": 2,
                        "num_data_workers": 0, "pin_memory": False},
        encoder_config={"name": "auto",
                        "pretrained_model_name_or_path": "bert-tiny",
                        "half_precision": False},
        pooler_config={"name": "mean"},
        embedder_config={"name": "full_sequence"},
        writer_config={"name": "huggingface"},
    )
    main(cfg)
    shards = list((odir / "embeddings").iterdir())
    assert len(shards) == 2
    assert (odir / "config.yaml").exists()


def test_semantic_chunk_all_buffers_filtered(tmp_path):
    """Docs whose sentences are all shorter than min_buffer_length must
    produce an EMPTY result, not an IndexError crash — one short file
    must not kill a distributed embed run."""
    import json

    from distllm_amd.embed.datasets import (JsonlChunkDataset,
                                            JsonlChunkDatasetConfig)
    from distllm_amd.embed.embedders import (SemanticChunkEmbedder,
                                             SemanticChunkEmbedderConfig)
    from distllm_amd.embed import get_encoder, get_pooler

    f = tmp_path / "short.jsonl"
    f.write_text(json.dumps({"text": "Too short. Really.",
                             "path": "p0"}) + "\n")
    enc = get_encoder({"name": "auto",
                       "pretrained_model_name_or_path": "bert-tiny",
                       "half_precision": False})
    pooler = get_pooler({"name": "mean"})
    ds = JsonlChunkDataset(JsonlChunkDatasetConfig())
    loader = ds.get_dataloader(f, enc)
    emb = SemanticChunkEmbedder(SemanticChunkEmbedderConfig())
    result = emb.embed(loader, enc, pooler)
    assert result.embeddings.shape[0] == 0
    assert result.text == []


def test_hf_writer_skips_empty_result(tmp_path):
    import numpy as np

    from distllm_amd.embed.embedders import EmbedderResult
    from distllm_amd.embed.writers import get_writer

    w = get_writer({"name": "huggingface"})
    out = tmp_path / "shard"
    w.write(out, EmbedderResult(
        embeddings=np.zeros((0, 0), dtype=np.float32), text=[],
        metadata=[]))
    # nothing written: an empty saved HF dataset cannot be loaded back
    assert not out.exists() or not any(out.iterdir())
