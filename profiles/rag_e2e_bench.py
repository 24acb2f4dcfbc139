"""End-to-end RAG question-answering throughput (BASELINE config #5):
HIP flat-index kNN retrieval (PubMedBERT query encoder) + Llama-3-8B
generation, measured as questions/second.

Reference anchor: the reference's MCQA pipeline measured
~5 questions/s on a single A100-class GPU with individual requests and
~60-100 q/s on 8 GPUs with vLLM TP + client batching
(MULTI_GPU_BATCHING_GUIDE.md:164-169). Synthetic corpus + random-init
weights (no network), answer length capped like MCQA answers.

  python profiles/rag_e2e_bench.py --questions 64 --corpus 100000
"""

import os
os.environ.setdefault("DISTLLM_ALLOW_HASH_TOKENIZER", "1")
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import numpy as np
import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--questions", type=int, default=64)
    ap.add_argument("--corpus", type=int, default=100_000)
    ap.add_argument("--top-k", type=int, default=5)
    ap.add_argument("--max-tokens", type=int, default=64)
    ap.add_argument("--gen-model", default="llama3-8b")
    ap.add_argument("--workdir", type=Path, default=Path("/tmp/rag_e2e"))
    args = ap.parse_args()

    from datasets import Dataset as HFDataset

    from distllm_amd.embed import get_encoder, get_pooler
    from distllm_amd.generate import get_generator, get_prompt_template
    from distllm_amd.rag import RagGenerator, Retriever
    from distllm_amd.rag.search import HipFlatIndex, normalize_L2

    t0 = time.perf_counter()
    rng = np.random.default_rng(0)
    emb = rng.standard_normal((args.corpus, 768)).astype(np.float32)
    normalize_L2(emb)
    corpus_dir = args.workdir / "corpus"
    if not corpus_dir.exists():
        HFDataset.from_dict({
            "text": [f"Scientific finding number {i}: lorem ipsum "
                     f"dolor sit amet variant {i % 97}." for i in
                     range(args.corpus)],
            "embeddings": [e for e in emb],
        }).save_to_disk(str(corpus_dir))
    index = HipFlatIndex(corpus_dir, args.workdir / "index.npz")
    enc = get_encoder({"name": "auto",
                       "pretrained_model_name_or_path": "pubmedbert-base"})
    pooler = get_pooler({"name": "mean"})
    retriever = Retriever(encoder=enc, pooler=pooler, faiss_index=index,
                          batch_size=16)
    gen = get_generator({"name": "native", "model": args.gen_model,
                         "temperature": 0.5, "min_p": 0.1,
                         "max_tokens": args.max_tokens,
                         "max_batch_size": 128, "max_model_len": 2048})
    rag = RagGenerator(generator=gen, retriever=retriever)
    tmpl = get_prompt_template({"name": "question_answer"})
    print(f"setup in {time.perf_counter() - t0:.1f}s "
          f"(corpus {args.corpus}, gen {args.gen_model})")

    questions = [
        f"What does scientific finding number {i * 13 % args.corpus} "
        f"imply about variant {i % 97}?\nOptions:\n1. alpha\n2. beta\n"
        f"3. gamma\n4. delta\n"
        for i in range(args.questions)
    ]
    # warmup (3 questions, triggers autotune + graph capture)
    rag.generate(questions[:3], prompt_template=tmpl,
                 retrieval_top_k=args.top_k)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    answers = rag.generate(questions, prompt_template=tmpl,
                           retrieval_top_k=args.top_k)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = time.perf_counter() - t1
    qps = args.questions / dt
    print(f"RAG e2e: {args.questions} questions in {dt:.2f}s = "
          f"{qps:.2f} questions/s (retrieval top-{args.top_k} over "
          f"{args.corpus} chunks + {args.gen_model} generate "
          f"<= {args.max_tokens} tokens)  [reference anchor: ~5 q/s "
          f"single GPU, ~60-100 q/s on 8 GPUs]")
    assert len(answers) == args.questions


if __name__ == "__main__":
    main()
