"""Dataset strategies for the embed pipeline.

Capability parity with reference distllm/embed/datasets/* (jsonl,
jsonl_chunk, fasta, sequence_per_line, huggingface + InMemoryDataset /
DataCollator), same YAML `name:` discriminators and fields.
"""

from __future__ import annotations

import json
import re
from dataclasses import dataclass
from pathlib import Path
from typing import Any, Literal, Union

from pydantic import Field
from torch.utils.data import DataLoader, Dataset as TorchDataset

from ..utils import BaseConfig
from .splitter import sentences_to_buffers, split_sentences


class InMemoryDataset(TorchDataset):
    """In-RAM list dataset with optional per-item metadata."""

    def __init__(self, data: list[str], metadata: list[dict] | None = None):
        self.data = data
        self.metadata = metadata

    def __len__(self) -> int:
        return len(self.data)

    def __getitem__(self, idx: int) -> str:
        return self.data[idx]


class DataCollator:
    """Tokenize a batch of texts (padding + truncation to model max)."""

    def __init__(self, tokenizer, pad_to_multiple_of: int | None = 8):
        self.tokenizer = tokenizer
        self.pad_to_multiple_of = pad_to_multiple_of

    def __call__(self, batch: list[str]):
        return self.tokenizer(
            batch,
            padding=True,
            truncation=True,
            return_tensors="pt",
            pad_to_multiple_of=self.pad_to_multiple_of,
        )


# ---------------------------------------------------------------------------
# fasta support
# ---------------------------------------------------------------------------

@dataclass
class Sequence:
    sequence: str
    tag: str


_FASTA_RE = re.compile(r"^>", re.MULTILINE)


def read_fasta(path: Path | str) -> list[Sequence]:
    text = Path(path).read_text()
    out: list[Sequence] = []
    for block in _FASTA_RE.split(text):
        if not block.strip():
            continue
        lines = block.splitlines()
        tag = lines[0].strip()
        seq = "".join(x.strip() for x in lines[1:] if not x.startswith(">"))
        if seq:
            out.append(Sequence(sequence=seq.upper(), tag=tag))
    return out


def write_fasta(seqs: list[Sequence], path: Path | str) -> None:
    with open(path, "w") as fh:
        for s in seqs:
            fh.write(f">{s.tag}\n{s.sequence}\n")


# ---------------------------------------------------------------------------
# dataset strategies
# ---------------------------------------------------------------------------

class _LoaderMixin:
    def _make_loader(self, data, metadata, encoder):
        cfg = self.config
        return DataLoader(
            pin_memory=cfg.pin_memory,
            batch_size=cfg.batch_size,
            num_workers=cfg.num_data_workers,
            dataset=InMemoryDataset(data, metadata),
            collate_fn=DataCollator(encoder.tokenizer),
        )


class JsonlDatasetConfig(BaseConfig):
    name: Literal["jsonl"] = "jsonl"
    text_field: str = "text"
    num_data_workers: int = 4
    batch_size: int = 8
    pin_memory: bool = True


class JsonlDataset(_LoaderMixin):
    def __init__(self, config: JsonlDatasetConfig):
        self.config = config

    def get_dataloader(self, data_file: Path, encoder) -> DataLoader:
        lines = Path(data_file).read_text().strip().split("\n")
        content = [json.loads(ln) for ln in lines if ln.strip()]
        data = [item.pop(self.config.text_field) for item in content]
        metadata = content if all(content) else None
        return self._make_loader(data, metadata, encoder)


class JsonlChunkDatasetConfig(BaseConfig):
    name: Literal["jsonl_chunk"] = "jsonl_chunk"
    text_field: str = "text"
    num_data_workers: int = 4
    batch_size: int = 8
    pin_memory: bool = True
    min_buffer_length: int = Field(
        default=750,
        description="buffers with fewer characters are filtered out",
    )
    buffer_size: int = Field(
        default=1,
        description="sentences grouped on each side of the focal sentence",
    )


class JsonlChunkDataset(_LoaderMixin):
    """Sentence-windowed buffers for semantic chunking (reference
    embed/datasets/jsonl_chunk.py:97-179 semantics)."""

    def __init__(self, config: JsonlChunkDatasetConfig):
        self.config = config

    def get_dataloader(self, data_file: Path, encoder) -> DataLoader:
        lines = Path(data_file).read_text().strip().split("\n")
        content = [json.loads(ln) for ln in lines if ln.strip()]
        data = [item.pop(self.config.text_field) for item in content]
        metadata = content
        if not metadata or any(not item for item in metadata):
            raise ValueError(
                "Metadata is empty; jsonl_chunk requires at least one "
                "non-text field (e.g. path) per line."
            )
        buffers: list[str] = []
        metadatas: list[dict] = []
        for idx, text in enumerate(data):
            sentences = split_sentences(text)
            bufs = sentences_to_buffers(sentences, self.config.buffer_size)
            buffers.extend(bufs)
            for sentence in sentences:
                md = metadata[idx].copy()
                md["sentence"] = sentence
                metadatas.append(md)
        keep = [
            i
            for i, buf in enumerate(buffers)
            if len(buf) > self.config.min_buffer_length
        ]
        buffers = [buffers[i] for i in keep]
        metadatas = [metadatas[i] for i in keep]
        return self._make_loader(buffers, metadatas, encoder)


class FastaDatasetConfig(BaseConfig):
    name: Literal["fasta"] = "fasta"
    num_data_workers: int = 4
    batch_size: int = 8
    pin_memory: bool = True


class FastaDataset(_LoaderMixin):
    def __init__(self, config: FastaDatasetConfig):
        self.config = config

    def get_dataloader(self, data_file: Path, encoder) -> DataLoader:
        seqs = read_fasta(data_file)
        data = [s.sequence for s in seqs]
        metadata = [
            {"tags": s.tag, "paths": str(data_file)} for s in seqs
        ]
        return self._make_loader(data, metadata, encoder)


class SequencePerLineDatasetConfig(BaseConfig):
    name: Literal["sequence_per_line"] = "sequence_per_line"
    num_data_workers: int = 4
    batch_size: int = 8
    pin_memory: bool = True
    header_lines: int = 0


class SequencePerLineDataset(_LoaderMixin):
    def __init__(self, config: SequencePerLineDatasetConfig):
        self.config = config

    def get_dataloader(self, data_file: Path, encoder) -> DataLoader:
        lines = Path(data_file).read_text().splitlines()
        data = [x for x in lines[self.config.header_lines :] if x.strip()]
        return self._make_loader(data, None, encoder)


class HuggingFaceDatasetConfig(BaseConfig):
    name: Literal["huggingface"] = "huggingface"
    text_field: str = "text"
    metadata_fields: list[str] = Field(default_factory=list)
    num_data_workers: int = 4
    batch_size: int = 8
    pin_memory: bool = True


class HuggingFaceDataset(_LoaderMixin):
    def __init__(self, config: HuggingFaceDatasetConfig):
        self.config = config

    def get_dataloader(self, data_file: Path, encoder) -> DataLoader:
        from datasets import Dataset as HFDataset

        ds = HFDataset.load_from_disk(str(data_file))
        data = list(ds[self.config.text_field])
        metadata = None
        if self.config.metadata_fields:
            cols = {f: ds[f] for f in self.config.metadata_fields}
            metadata = [
                {f: cols[f][i] for f in self.config.metadata_fields}
                for i in range(len(ds))
            ]
        return self._make_loader(data, metadata, encoder)


DatasetConfigs = Union[
    JsonlDatasetConfig,
    JsonlChunkDatasetConfig,
    FastaDatasetConfig,
    SequencePerLineDatasetConfig,
    HuggingFaceDatasetConfig,
]

STRATEGIES: dict[str, tuple[type, type]] = {
    "jsonl": (JsonlDatasetConfig, JsonlDataset),
    "jsonl_chunk": (JsonlChunkDatasetConfig, JsonlChunkDataset),
    "fasta": (FastaDatasetConfig, FastaDataset),
    "sequence_per_line": (SequencePerLineDatasetConfig,
                          SequencePerLineDataset),
    "huggingface": (HuggingFaceDatasetConfig, HuggingFaceDataset),
}


def get_dataset(kwargs: dict[str, Any]):
    name = kwargs.get("name", "")
    if name not in STRATEGIES:
        raise ValueError(
            f"unknown dataset name {name!r}; options: {sorted(STRATEGIES)}"
        )
    cfg_cls, impl_cls = STRATEGIES[name]
    return impl_cls(cfg_cls(**kwargs))
