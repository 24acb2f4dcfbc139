"""Prefill-throughput and IVF-vs-exact search probes (GPU evidence).

    python profiles/prefill_ivf_probe.py [--skip-prefill]
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import numpy as np
import torch


def bench_prefill():
    from distllm_amd.engine import LLMEngine, SamplingParams
    from distllm_amd.engine.engine import _Seq
    from distllm_amd.models import build_model

    torch.manual_seed(0)
    model = build_model("llama3-8b", device="cuda:0", dtype=torch.bfloat16)
    engine = LLMEngine(model, max_batch_size=32, max_model_len=2048 + 64)
    params = SamplingParams(temperature=0.0, min_p=0.0, max_tokens=4,
                            stop_token_ids=())

    def one(B, P):
        prompts = torch.randint(4, model.dims.vocab_size, (B, P)).tolist()
        seqs = [_Seq(seq_id=i, prompt_ids=p) for i, p in enumerate(prompts)]
        for s in seqs:
            s.blocks = engine.blockman.alloc(engine.blockman.blocks_for(P + 8))
        res = {}
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        engine._run_prefill(seqs, params, res)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        for s in seqs:
            engine.blockman.release(s.blocks)
        return B * P / dt

    for B, P in [(32, 512), (32, 1024), (16, 2048)]:
        one(B, P)
        r = sum(one(B, P) for _ in range(3)) / 3
        print(f"prefill B={B} P={P}: {r / 1000:.1f}k tok/s")


def bench_ivf():
    import tempfile

    from datasets import Dataset as HFDataset

    from distllm_amd.rag.search import HipFlatIndex

    rng = np.random.default_rng(0)
    N, D, Q = 1_000_000, 384, 256
    # clustered corpus (mixture of Gaussians): real embedding spaces are
    # clustered, which is the structure IVF exploits; uniform random data
    # would make any IVF look artificially bad.
    centers = rng.standard_normal((1024, D), dtype=np.float32)
    assign = rng.integers(0, 1024, N)
    emb = centers[assign] + 0.35 * rng.standard_normal((N, D),
                                                       dtype=np.float32)
    emb /= np.linalg.norm(emb, axis=1, keepdims=True)
    queries = emb[rng.integers(0, N, Q)] \
        + 0.05 * rng.standard_normal((Q, D), dtype=np.float32)

    tmp = Path(tempfile.mkdtemp())
    HFDataset.from_dict({
        "text": [""] * N, "embeddings": emb,
    }).save_to_disk(str(tmp / "ds"))

    t0 = time.perf_counter()
    idx_e = HipFlatIndex(tmp / "ds", tmp / "exact.npz")
    print(f"exact build: {time.perf_counter() - t0:.1f}s")
    r = idx_e.search(queries, top_k=10)
    t0 = time.perf_counter()
    r = idx_e.search(queries, top_k=10)
    te = time.perf_counter() - t0
    ie = r.total_indices

    t0 = time.perf_counter()
    idx_i = HipFlatIndex(tmp / "ds", tmp / "ivf.npz",
                         search_algorithm="ivf", nlist=1024, nprobe=32)
    print(f"ivf build: {time.perf_counter() - t0:.1f}s")
    r = idx_i.search(queries, top_k=10)
    t0 = time.perf_counter()
    r = idx_i.search(queries, top_k=10)
    ti = time.perf_counter() - t0
    ii = r.total_indices

    rec = np.mean([len(set(list(a)) & set(list(b))) / 10
                   for a, b in zip(ie, ii)])
    print(f"exact 1M x {Q}q k=10: {te * 1000:.1f} ms ({Q / te:.0f} q/s)")
    print(f"ivf nlist=1024 nprobe=32: {ti * 1000:.1f} ms ({Q / ti:.0f} q/s) "
          f"recall@10 vs exact = {rec:.3f}")

    # single-query latency (the chat/serving path)
    q1 = queries[:1]
    for name, idx in (("exact", idx_e), ("ivf", idx_i)):
        idx.search(q1, top_k=10)
        t0 = time.perf_counter()
        for _ in range(50):
            idx.search(q1, top_k=10)
        dt = (time.perf_counter() - t0) / 50
        print(f"{name} single-query latency: {dt * 1e3:.2f} ms "
              f"({1 / dt:.0f} q/s single-stream)")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--skip-prefill", action="store_true")
    a = ap.parse_args()
    if not a.skip_prefill:
        bench_prefill()
    bench_ivf()
