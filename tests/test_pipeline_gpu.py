"""GPU end-to-end pipeline tests: the real embed path (HIP kernels)
on jsonl + fasta fixtures, and retrieval on GPU."""

import json

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _write_jsonl(path, n=6):
    with open(path, "w") as fh:
        for i in range(n):
            text = " ".join(
                f"Sentence {j} of document {i} has enough words to count."
                for j in range(30)
            )
            fh.write(json.dumps({"text": text, "path": f"doc{i}"}) + "\n")


def test_embed_semantic_chunk_gpu(tmp_path, device):
    from distllm_amd.embed import (
        get_dataset,
        get_embedder,
        get_encoder,
        get_pooler,
    )

    f = tmp_path / "a.jsonl"
    _write_jsonl(f, n=4)
    enc = get_encoder({"name": "auto",
                       "pretrained_model_name_or_path": "pubmedbert-base"})
    assert str(enc.device).startswith("cuda")
    assert enc.dtype == torch.bfloat16
    ds = get_dataset({"name": "jsonl_chunk", "batch_size": 16,
                      "num_data_workers": 0, "min_buffer_length": 20})
    pooler = get_pooler({"name": "mean"})
    embedder = get_embedder({"name": "semantic_chunk",
                             "chunk_batch_size": 16,
                             "min_chunk_length": 10,
                             "normalize_embeddings": True})
    result = embedder.embed(ds.get_dataloader(f, enc), enc, pooler)
    assert result.embeddings.shape[0] > 0
    assert np.isfinite(result.embeddings).all()
    norms = np.linalg.norm(result.embeddings, axis=1)
    assert np.allclose(norms, 1.0, atol=1e-2)


def test_embed_fasta_esm2_gpu(tmp_path, device):
    from distllm_amd.embed import (
        get_dataset,
        get_embedder,
        get_encoder,
        get_pooler,
    )

    f = tmp_path / "p.fasta"
    seqs = ["MKVLATGLYQWE" * 6, "ACDEFGHIKLMNPQRSTVWY" * 3, "MKGV" * 20]
    f.write_text("".join(f">s{i}\n{s}\n" for i, s in enumerate(seqs)))
    enc = get_encoder({"name": "esm2",
                       "pretrained_model_name_or_path": "esm2-8m"})
    ds = get_dataset({"name": "fasta", "batch_size": 2,
                      "num_data_workers": 0})
    pooler = get_pooler({"name": "mean"})
    embedder = get_embedder({"name": "full_sequence",
                             "normalize_embeddings": True})
    result = embedder.embed(ds.get_dataloader(f, enc), enc, pooler)
    assert result.embeddings.shape == (3, 320)
    assert np.isfinite(result.embeddings).all()


def test_search_index_gpu(tmp_path, device):
    from datasets import Dataset as HFDataset

    from distllm_amd.rag.search import HipFlatIndex

    rng = np.random.default_rng(5)
    emb = rng.standard_normal((5000, 128)).astype(np.float32)
    emb /= np.linalg.norm(emb, axis=1, keepdims=True)
    d = tmp_path / "c"
    HFDataset.from_dict({
        "text": [f"t{i}" for i in range(5000)],
        "embeddings": [e for e in emb],
    }).save_to_disk(str(d))
    idx = HipFlatIndex(d, tmp_path / "i.npz")
    assert str(idx.corpus.device).startswith("cuda")
    res = idx.search(emb[[11, 4321]], top_k=10)
    assert res.total_indices[0][0] == 11
    assert res.total_indices[1][0] == 4321
    # ubinary path on GPU
    bidx = HipFlatIndex(d, tmp_path / "b.npz", precision="ubinary",
                        rescore_multiplier=4)
    res2 = bidx.search(emb[[99]], top_k=5)
    assert res2.total_indices[0][0] == 99


def test_mistral_embedder_gpu(device):
    """SFR-Mistral-style last-token embed path on GPU (tiny dims)."""
    from distllm_amd.embed import get_encoder, get_pooler

    enc = get_encoder({"name": "auto",
                       "pretrained_model_name_or_path": "mistral-tiny"})
    pooler = get_pooler({"name": "last_token"})
    tok = enc.tokenizer(["Embed this text", "And this longer one too"],
                        return_tensors="pt", pad_to_multiple_of=8)
    hidden = enc.encode(tok)
    pooled = pooler.pool(hidden,
                         tok["attention_mask"].to(hidden.device),
                         normalize=True)
    assert pooled.shape == (2, 256)
    assert torch.isfinite(pooled).all()
