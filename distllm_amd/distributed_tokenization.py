"""Distributed tokenization entry point.

Reference parity: distllm/distributed_tokenization.py (jsonl -> tokenizer
-> HF dataset {input_ids, attention_mask[, labels]}), Parsl replaced by
the per-rank shard.
"""

from __future__ import annotations

import argparse
import json
from pathlib import Path
from typing import Any

from .embed.tokenizers import load_tokenizer
from .embed.writers import unique_shard_dir
from .parallel import barrier, init_distributed, shard_list
from .utils import BaseConfig, Timer


class TokenizationConfig(BaseConfig):
    input_dir: Path
    output_dir: Path
    glob_patterns: list[str] = ["*.jsonl"]
    text_field: str = "text"
    tokenizer_name: str = "pubmedbert-base"
    tokenizer_path: str = ""
    max_length: int = 512
    save_labels: bool = False
    compute_config: dict[str, Any] = {"name": "local"}


def tokenizer_worker(input_path: Path, output_dir: Path,
                     config: TokenizationConfig) -> Path:
    from datasets import Dataset as HFDataset

    tok = load_tokenizer(config.tokenizer_path or config.tokenizer_name,
                         model_max_length=config.max_length)
    texts = []
    for line in Path(input_path).read_text().splitlines():
        if line.strip():
            texts.append(json.loads(line)[config.text_field])
    enc = tok(texts, padding=True, truncation=True,
              max_length=config.max_length, return_tensors="pt")
    cols = {
        "input_ids": enc["input_ids"].tolist(),
        "attention_mask": enc["attention_mask"].tolist(),
    }
    if config.save_labels:
        cols["labels"] = cols["input_ids"]
    ds = HFDataset.from_dict(cols)
    shard = unique_shard_dir(output_dir)
    ds.save_to_disk(str(shard))
    return shard


def main(config: TokenizationConfig) -> None:
    rank, world = init_distributed()
    config.output_dir.mkdir(parents=True, exist_ok=True)
    if rank == 0:
        config.write_yaml(config.output_dir / "config.yaml")
    files: list[Path] = []
    for pattern in config.glob_patterns:
        files.extend(sorted(config.input_dir.glob(pattern)))
    out = config.output_dir / "tokenized"
    out.mkdir(exist_ok=True, parents=True)
    for path in shard_list(files):
        with Timer("finished-tokenization", str(path), f"rank{rank}"):
            tokenizer_worker(path, out, config)
    barrier()


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--config", type=Path, required=True)
    args = parser.parse_args()
    main(TokenizationConfig.from_yaml(args.config))
