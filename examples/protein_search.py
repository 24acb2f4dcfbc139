"""End-to-end protein semantic search (reference
examples/protein_search.py): embed a fasta corpus with ESM2/ESMC +
mean pooling, build a HIP flat index (ubinary by default), query it.

  python examples/protein_search.py --fasta seqs.fasta \
      --model esm2-650m --query MKVLAT... --top-k 5
"""

from __future__ import annotations

import argparse
import tempfile
from pathlib import Path


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--fasta", type=Path, required=True)
    ap.add_argument("--model", default="esm2-8m")
    ap.add_argument("--encoder", default="esm2", choices=["esm2", "esmc"])
    ap.add_argument("--precision", default="ubinary",
                    choices=["float32", "ubinary"])
    ap.add_argument("--query", required=True, help="protein sequence")
    ap.add_argument("--top-k", type=int, default=5)
    ap.add_argument("--batch-size", type=int, default=8)
    ap.add_argument("--workdir", type=Path, default=None)
    args = ap.parse_args()

    from datasets import Dataset as HFDataset

    from distllm_amd.embed import get_encoder, get_pooler
    from distllm_amd.embed.datasets import read_fasta
    from distllm_amd.rag.search import HipFlatIndex, Retriever

    work = args.workdir or Path(tempfile.mkdtemp(prefix="protein_search_"))
    enc = get_encoder({
        "name": args.encoder,
        "pretrained_model_name_or_path": args.model,
    })
    pooler = get_pooler({"name": "mean"})

    seqs = read_fasta(args.fasta)
    print(f"embedding {len(seqs)} sequences from {args.fasta}")
    embs = []
    import torch

    for i in range(0, len(seqs), args.batch_size):
        batch = [s.sequence for s in seqs[i : i + args.batch_size]]
        tok = enc.tokenizer(batch, padding=True, truncation=True,
                            return_tensors="pt", pad_to_multiple_of=8)
        with torch.no_grad():
            hidden = enc.encode(tok)
            pooled = pooler.pool(
                hidden, tok["attention_mask"].to(hidden.device),
                normalize=True)
        embs.append(pooled.float().cpu())
    emb = torch.cat(embs).numpy()

    corpus_dir = work / "corpus"
    HFDataset.from_dict({
        "text": [s.sequence for s in seqs],
        "tags": [s.tag for s in seqs],
        "embeddings": [e for e in emb],
    }).save_to_disk(str(corpus_dir))

    index = HipFlatIndex(corpus_dir, work / "index.npz",
                         precision=args.precision, rescore_multiplier=4)
    retriever = Retriever(encoder=enc, pooler=pooler, faiss_index=index,
                          batch_size=args.batch_size)
    results, _ = retriever.search([args.query], top_k=args.top_k)
    tags = retriever.get(results.total_indices[0], "tags")
    print(f"\ntop-{args.top_k} matches for query:")
    for rank, (idx, score, tag) in enumerate(zip(
            results.total_indices[0], results.total_scores[0], tags)):
        print(f"  {rank + 1}. [{score:.4f}] {tag}")


if __name__ == "__main__":
    main()
