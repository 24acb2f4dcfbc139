"""CPU tests of the RAG core: index build/save/load, exact + ubinary
search vs numpy reference, retriever, end-to-end RagGenerator."""

import numpy as np
import pytest

from distllm_amd.rag.search import (
    HipFlatIndex,
    Retriever,
    normalize_L2,
    quantize_embeddings,
)


@pytest.fixture()
def corpus_dataset(tmp_path):
    from datasets import Dataset as HFDataset

    rng = np.random.default_rng(3)
    emb = rng.standard_normal((50, 32)).astype(np.float32)
    emb = emb / np.linalg.norm(emb, axis=1, keepdims=True)
    ds = HFDataset.from_dict({
        "text": [f"doc {i}" for i in range(50)],
        "embeddings": [e for e in emb],
    })
    d = tmp_path / "corpus"
    ds.save_to_disk(str(d))
    return d, emb


def test_exact_search_matches_numpy(tmp_path, corpus_dataset):
    d, emb = corpus_dataset
    idx = HipFlatIndex(d, tmp_path / "idx.npz")
    q = emb[[3, 17]] + 0.01
    res = idx.search(q, top_k=5)
    ref = np.argsort(-(q @ emb.T), axis=1)[:, :5]
    assert res.total_indices[0][0] == 3
    assert res.total_indices[1][0] == 17
    for got, want in zip(res.total_indices, ref):
        assert set(got) == set(want.tolist())


def test_index_save_load(tmp_path, corpus_dataset):
    d, emb = corpus_dataset
    p = tmp_path / "idx.npz"
    HipFlatIndex(d, p)
    assert p.exists()
    idx2 = HipFlatIndex(d, p)  # load path
    res = idx2.search(emb[[5]], top_k=1)
    assert res.total_indices[0][0] == 5
    assert idx2.get([5], "text") == ["doc 5"]


def test_ubinary_search(tmp_path, corpus_dataset):
    d, emb = corpus_dataset
    idx = HipFlatIndex(d, tmp_path / "bidx.npz", precision="ubinary",
                       rescore_multiplier=4)
    res = idx.search(emb[[7, 30]], top_k=3)
    assert res.total_indices[0][0] == 7
    assert res.total_indices[1][0] == 30


def test_score_threshold_filter(tmp_path, corpus_dataset):
    d, emb = corpus_dataset
    idx = HipFlatIndex(d, tmp_path / "idx.npz")
    res = idx.search(emb[[0]], top_k=10, score_threshold=0.99)
    # only the exact match should survive a 0.99 cosine threshold
    assert res.total_indices[0] == [0]


def test_quantize_roundtrip():
    emb = np.array([[0.5, -0.2, 0.1, -0.9, 0.3, 0.2, -0.1, 0.4]],
                   dtype=np.float32)
    q = quantize_embeddings(emb, "ubinary")
    assert q.dtype == np.uint8 and q.shape == (1, 1)
    bits = np.unpackbits(q, axis=1)[0]
    assert list(bits) == [1, 0, 1, 0, 1, 1, 0, 1]


def test_retriever_and_rag_generator(tmp_path):
    """End-to-end: embed corpus with the real encoder, search with the
    same encoder, generate with the native engine."""
    from datasets import Dataset as HFDataset

    from distllm_amd.embed import get_encoder, get_pooler
    from distllm_amd.generate import get_generator, get_prompt_template
    from distllm_amd.rag import RagGenerator

    enc = get_encoder({"name": "auto",
                       "pretrained_model_name_or_path": "bert-tiny",
                       "half_precision": False})
    pooler = get_pooler({"name": "mean"})
    texts = [f"Document number {i} about topic {i % 3}." for i in range(9)]
    tok = enc.tokenizer(texts, return_tensors="pt")
    hidden = enc.encode(tok)
    pooled = pooler.pool(hidden, tok["attention_mask"], normalize=True)
    emb = pooled.float().cpu().numpy()
    ds = HFDataset.from_dict({"text": texts, "embeddings": [e for e in emb]})
    d = tmp_path / "c"
    ds.save_to_disk(str(d))

    idx = HipFlatIndex(d, tmp_path / "i.npz")
    retr = Retriever(encoder=enc, pooler=pooler, faiss_index=idx,
                     batch_size=2)
    res, qemb = retr.search("Document number 4 about topic 1.", top_k=3)
    assert len(res.total_indices[0]) == 3
    assert 4 in res.total_indices[0]
    assert retr.get_texts([0]) == [texts[0]]

    gen = get_generator({"name": "native", "model": "llama-tiny",
                         "temperature": 0.0, "min_p": 0.0,
                         "max_tokens": 4, "max_model_len": 256})
    rag = RagGenerator(generator=gen, retriever=retr)
    out = rag.generate(["What is topic 1?"],
                       prompt_template=get_prompt_template(
                           {"name": "question_answer"}),
                       retrieval_top_k=2)
    assert len(out) == 1 and isinstance(out[0], str)


def test_ivf_search(tmp_path):
    from datasets import Dataset as HFDataset

    rng = np.random.default_rng(9)
    # clustered corpus: 20 centers x 40 points
    centers = rng.standard_normal((20, 24)).astype(np.float32) * 3
    emb = np.concatenate([
        c + 0.1 * rng.standard_normal((40, 24)).astype(np.float32)
        for c in centers
    ])
    emb = normalize_L2(emb)
    d = tmp_path / "c"
    HFDataset.from_dict({
        "text": [f"t{i}" for i in range(len(emb))],
        "embeddings": [e for e in emb],
    }).save_to_disk(str(d))
    idx = HipFlatIndex(d, tmp_path / "i.npz", search_algorithm="ivf",
                       nlist=20, nprobe=4)
    q = emb[[5, 333, 777]]
    res = idx.search(q, top_k=5)
    for qi, want in zip(range(3), [5, 333, 777]):
        assert res.total_indices[qi][0] == want
    # recall vs exact on clustered data should be high
    exact = HipFlatIndex(d, tmp_path / "e.npz")
    ref = exact.search(q, top_k=5)
    overlap = sum(
        len(set(a) & set(b))
        for a, b in zip(res.total_indices, ref.total_indices)
    ) / 15.0
    assert overlap >= 0.8, overlap


def test_run_eval_suite_end_to_end(tmp_path):
    """Full eval-suite run offline: fixture task data pre-placed (skips
    curl), native tiny generator, metrics written to the results JSON
    (reference rag/evaluate.py:18-99 + tasks)."""
    import json

    from distllm_amd.rag.evaluate import EvalSuiteConfig, run_eval_suite

    dl = tmp_path / "eval_data"
    (dl / "sciq").mkdir(parents=True)
    (dl / "sciq" / "test.json").write_text(json.dumps([{
        "question": "What is water",
        "correct_answer": "H2O",
        "distractor1": "CO2", "distractor2": "O2", "distractor3": "N2",
    }] * 3))
    cfg = EvalSuiteConfig(
        rag_configs=[{
            "generator_config": {"name": "native", "model": "llama-tiny",
                                 "max_tokens": 4, "max_batch_size": 4,
                                 "max_model_len": 256},
        }],
        tasks=["sciq"],
        download_dir=dl,
        output_path=tmp_path / "results.json",
    )
    rows = run_eval_suite(cfg)
    assert len(rows) == 1
    assert rows[0]["task"] == "sciq" and 0.0 <= rows[0]["accuracy"] <= 1.0
    saved = json.loads((tmp_path / "results.json").read_text())
    assert saved == rows


def test_reference_evaluate_yaml_shape_parses():
    """A reference examples/evaluate/*.yaml-shaped config (V1 faiss
    block, vllm llm_name dialect, nested retriever tree) validates."""
    import yaml

    from distllm_amd.rag.evaluate import EvalSuiteConfig

    raw = yaml.safe_load("""
rag_configs:
  - generator_config:
      name: vllm
      llm_name: some/model
      top_p: 0.95
      tensor_parallel_size: 8
    retriever_config:
      faiss_config:
        name: faiss_index_v1
        dataset_dir: /data/merged
      encoder_config:
        name: auto
        pretrained_model_name_or_path: pritamdeka/S-PubMedBert-MS-MARCO
      pooler_config:
        name: mean
tasks: [sciq, pubmedqa, litqa, protein_interaction_qa, protein_function_qa]
download_dir: /data/benchmarks
""")
    c = EvalSuiteConfig(**raw)
    assert len(c.rag_configs) == 1
    fc = c.rag_configs[0].retriever_config.faiss_config
    assert str(fc.faiss_index_path).endswith("hip_index.npz")
