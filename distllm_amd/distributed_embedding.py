"""Distributed embedding entry point.

Reference parity: distllm/distributed_embedding.py (same YAML schema:
input_dir, output_dir, glob_patterns, dataset_config, encoder_config,
pooler_config, embedder_config, writer_config, compute_config), with
the Parsl task farm replaced by the RCCL/xGMI per-rank shard
(parallel/runtime.py). Each input file is embedded by exactly one rank
and written to a fresh uuid4 dataset dir, so re-runs and the offline
`merge` workflow behave like the reference's.

Run:
  single GPU:  python -m distllm_amd.distributed_embedding --config c.yaml
  8 GPUs:      python -m torch.distributed.run --nproc-per-node 8 \
                 --master-addr 127.0.0.1 \
                 -m distllm_amd.distributed_embedding --config c.yaml
"""

from __future__ import annotations

import argparse
from pathlib import Path
from typing import Any

from .embed import (
    get_dataset,
    get_embedder,
    get_encoder,
    get_pooler,
    get_writer,
    unique_shard_dir,
)
from .parallel import barrier, init_distributed, shard_list
from .utils import BaseConfig, Timer


class EmbeddingConfig(BaseConfig):
    input_dir: Path
    output_dir: Path
    glob_patterns: list[str] = ["*"]
    dataset_config: dict[str, Any]
    encoder_config: dict[str, Any]
    pooler_config: dict[str, Any]
    embedder_config: dict[str, Any]
    writer_config: dict[str, Any]
    compute_config: dict[str, Any] = {"name": "local"}
    retries: int = 1  # per-file retries (reference parsl.py:85,130,197)


def embedding_worker(
    input_path: Path,
    output_dir: Path,
    dataset_kwargs: dict[str, Any],
    encoder_kwargs: dict[str, Any],
    pooler_kwargs: dict[str, Any],
    embedder_kwargs: dict[str, Any],
    writer_kwargs: dict[str, Any],
) -> Path:
    """Embed one input file end-to-end (reference
    distributed_embedding.py:23-80 structure)."""
    with Timer("loaded-encoder", str(input_path)):
        encoder = get_encoder(encoder_kwargs, register=True)
    dataset = get_dataset(dataset_kwargs)
    pooler = get_pooler(pooler_kwargs)
    embedder = get_embedder(embedder_kwargs)
    writer = get_writer(writer_kwargs)

    with Timer("loaded-dataset", str(input_path)):
        dataloader = dataset.get_dataloader(input_path, encoder)

    with Timer("computed-embeddings", str(input_path)):
        result = embedder.embed(dataloader, encoder, pooler)

    with Timer("wrote-embeddings", str(input_path)):
        shard = unique_shard_dir(output_dir)
        writer.write(shard, result)
    return shard


def main(config: EmbeddingConfig) -> None:
    rank, world = init_distributed()
    config.output_dir.mkdir(exist_ok=True, parents=True)
    if rank == 0:
        config.write_yaml(config.output_dir / "config.yaml")

    files: list[Path] = []
    for pattern in config.glob_patterns:
        files.extend(sorted(config.input_dir.glob(pattern)))
    my_files = shard_list(files)
    print(f"[rank {rank}/{world}] embedding {len(my_files)} of "
          f"{len(files)} files")

    emb_dir = config.output_dir / "embeddings"
    emb_dir.mkdir(exist_ok=True, parents=True)
    failed: list[str] = []
    for path in my_files:
        # per-file retry replaces Parsl task retries: a failed file only
        # loses itself (share-nothing), and the run reports it at the end
        for attempt in range(config.retries + 1):
            try:
                with Timer("finished-embedding", str(path), f"rank{rank}"):
                    embedding_worker(
                        path, emb_dir,
                        config.dataset_config, config.encoder_config,
                        config.pooler_config, config.embedder_config,
                        config.writer_config,
                    )
                break
            except Exception as e:
                print(f"[rank {rank}] {path} attempt {attempt + 1} "
                      f"failed: {e!r}")
                if attempt == config.retries:
                    failed.append(str(path))
    barrier()
    if failed:
        print(f"[rank {rank}] FAILED files: {failed}")
    if rank == 0:
        print(f"[done] {len(files)} files -> {emb_dir}")


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--config", type=Path, required=True)
    args = parser.parse_args()
    main(EmbeddingConfig.from_yaml(args.config))
