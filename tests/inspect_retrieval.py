"""Manual retrieval inspection utility (reference
tests/inspect_retrieval.py): load a retriever config, run queries from
stdin, print scored chunks.

  python tests/inspect_retrieval.py --config retriever.yaml --top-k 5
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", type=Path, required=True)
    ap.add_argument("--top-k", type=int, default=5)
    args = ap.parse_args()

    import yaml

    from distllm_amd.rag.search import RetrieverConfig

    cfg = RetrieverConfig(**yaml.safe_load(args.config.read_text()))
    retriever = cfg.get_retriever()
    for line in sys.stdin:
        q = line.strip()
        if not q:
            continue
        results, _ = retriever.search([q], top_k=args.top_k)
        for score, idx in zip(results.total_scores[0],
                              results.total_indices[0]):
            text = retriever.get_texts([idx])[0]
            print(f"[{score:.4f}] ({idx}) {text[:160]}")
        print("---")


if __name__ == "__main__":
    main()
