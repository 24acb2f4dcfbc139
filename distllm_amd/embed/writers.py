"""Writer strategies: HF-datasets on-disk format (the interchange format
the whole pipeline reads back — columns `text`, `embeddings`, flattened
metadata) and numpy. Reference parity:
distllm/embed/writers/{huggingface,numpy}.py incl. the uuid4 output
dirs and fault-tolerant merge.
"""

from __future__ import annotations

import uuid
from pathlib import Path
from typing import Any, Literal, Union

import numpy as np

from ..utils import BaseConfig
from .embedders import EmbedderResult


class HuggingFaceWriterConfig(BaseConfig):
    name: Literal["huggingface"] = "huggingface"
    # reference writers/huggingface.py:43 — dataset-writing worker
    # count; accepted for YAML drop-in (shards here are written by
    # their owning rank, one process each)
    num_proc: int | None = None


class HuggingFaceWriter:
    def __init__(self, config: HuggingFaceWriterConfig):
        self.config = config

    def write(self, output_dir: Path, result: EmbedderResult) -> None:
        from datasets import Dataset as HFDataset

        if len(result.text) == 0:
            # a 0-row saved HF dataset cannot be load_from_disk'd back;
            # write nothing (merge would have to skip it anyway)
            print(f"[write] {output_dir}: empty result, no shard written")
            return
        rows: dict[str, list] = {
            "text": result.text,
            "embeddings": [e for e in result.embeddings],
        }
        if result.metadata is not None and len(result.metadata) > 0:
            keys = set()
            for md in result.metadata:
                keys.update(md.keys())
            for k in sorted(keys):
                rows[k] = [md.get(k) for md in result.metadata]
        ds = HFDataset.from_dict(rows)
        output_dir = Path(output_dir)
        output_dir.mkdir(parents=True, exist_ok=True)
        ds.save_to_disk(str(output_dir))

    def merge(self, dataset_dirs: list[Path], output_dir: Path,
              num_proc: int = 1) -> None:
        from datasets import concatenate_datasets, load_from_disk

        parts = []
        for d in dataset_dirs:
            try:
                parts.append(load_from_disk(str(d)))
            except Exception as e:  # skip missing/corrupt shards
                print(f"[merge] skipping {d}: {e}")
        if not parts:
            raise ValueError("no readable dataset shards to merge")
        merged = concatenate_datasets(parts)
        # num_proc=1 still makes `datasets` spawn a 1-worker pool (which
        # re-imports __main__ — breaks under embedded interpreters and
        # wastes a fork); only parallelize for num_proc > 1.
        merged.save_to_disk(str(output_dir),
                            num_proc=num_proc if num_proc and num_proc > 1
                            else None)


class NumpyWriterConfig(BaseConfig):
    name: Literal["numpy"] = "numpy"


class NumpyWriter:
    def __init__(self, config: NumpyWriterConfig):
        self.config = config

    def write(self, output_dir: Path, result: EmbedderResult) -> None:
        output_dir = Path(output_dir)
        output_dir.mkdir(parents=True, exist_ok=True)
        np.save(output_dir / "embeddings.npy", result.embeddings)
        np.save(output_dir / "text.npy", np.array(result.text, dtype=object),
                allow_pickle=True)
        if result.metadata is not None:
            np.save(output_dir / "metadata.npy",
                    np.array(result.metadata, dtype=object),
                    allow_pickle=True)

    def merge(self, dataset_dirs: list[Path], output_dir: Path,
              num_proc: int = 1) -> None:
        embs, texts, metas = [], [], []
        have_meta = True
        for d in dataset_dirs:
            d = Path(d)
            if not (d / "embeddings.npy").exists():
                print(f"[merge] skipping {d}: missing embeddings.npy")
                continue
            embs.append(np.load(d / "embeddings.npy"))
            texts.append(np.load(d / "text.npy", allow_pickle=True))
            mp = d / "metadata.npy"
            if mp.exists():
                metas.append(np.load(mp, allow_pickle=True))
            else:
                have_meta = False
        out = Path(output_dir)
        out.mkdir(parents=True, exist_ok=True)
        np.save(out / "embeddings.npy", np.concatenate(embs))
        np.save(out / "text.npy", np.concatenate(texts), allow_pickle=True)
        if have_meta and metas:
            np.save(out / "metadata.npy", np.concatenate(metas),
                    allow_pickle=True)


def unique_shard_dir(base: Path) -> Path:
    """Per-task output dir (uuid4, reference
    distributed_embedding.py:72-73 idempotent-rerun semantics)."""
    return Path(base) / str(uuid.uuid4())


WriterConfigs = Union[HuggingFaceWriterConfig, NumpyWriterConfig]

STRATEGIES: dict[str, tuple[type, type]] = {
    "huggingface": (HuggingFaceWriterConfig, HuggingFaceWriter),
    "numpy": (NumpyWriterConfig, NumpyWriter),
}


def get_writer(kwargs: dict[str, Any]):
    name = kwargs.get("name", "")
    if name not in STRATEGIES:
        raise ValueError(
            f"unknown writer name {name!r}; options: {sorted(STRATEGIES)}"
        )
    cfg_cls, impl_cls = STRATEGIES[name]
    return impl_cls(cfg_cls(**kwargs))
