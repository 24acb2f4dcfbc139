"""Embedder strategies: full_sequence (the hot loop) and semantic_chunk.

Reference parity: distllm/embed/embedders/{full_sequence,semantic_chunk}.py.
MI355X-first differences:
  - the hot loop keeps the forward entirely on-GPU (HIP kernels) and
    lands pooled fp32 embeddings asynchronously into a pinned host
    buffer (H2D upload of the next batch overlaps the D2H of the
    previous one on the same stream order);
  - the semantic-chunk distance pass runs on-GPU (adjacent_cosine HIP
    kernel) instead of the reference's CPU numpy loop; the percentile
    breakpoints use the same numpy percentile semantics on the host.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Literal, Union

import numpy as np
import torch
from pydantic import Field
from torch.utils.data import DataLoader

from .. import ops
from ..utils import BaseConfig
from .datasets import DataCollator, InMemoryDataset


@dataclass
class EmbedderResult:
    embeddings: np.ndarray
    text: list[str]
    metadata: list[dict] | None = None


@torch.no_grad()
def compute_embeddings(dataloader: DataLoader, encoder, pooler,
                       normalize: bool = False) -> np.ndarray:
    """The embed hot loop (reference full_sequence.py:21-80):
    tokenized batch -> H2D -> encoder forward -> fused pool(+norm) ->
    async D2H into a pinned [N, H] fp32 buffer."""
    n = len(dataloader.dataset)
    H = encoder.embedding_size
    on_gpu = str(encoder.device).startswith("cuda")
    host = torch.empty(n, H, dtype=torch.float32,
                       pin_memory=on_gpu)
    idx = 0
    for batch in dataloader:
        ids = batch["input_ids"]
        bsz = ids.shape[0]
        hidden = encoder.encode(batch)  # [B,S,H] model dtype, on device
        mask = batch["attention_mask"].to(hidden.device, non_blocking=True)
        if hidden.dtype != torch.bfloat16 and on_gpu:
            hidden = hidden.to(torch.bfloat16)
        pooled = pooler.pool(hidden, mask, normalize=normalize)  # [B,H] f32
        host[idx : idx + bsz].copy_(pooled, non_blocking=True)
        idx += bsz
    if on_gpu:
        torch.cuda.synchronize()
    return host.numpy()


class FullSequenceEmbedderConfig(BaseConfig):
    name: Literal["full_sequence"] = "full_sequence"
    normalize_embeddings: bool = False


class FullSequenceEmbedder:
    def __init__(self, config: FullSequenceEmbedderConfig):
        self.config = config

    def embed(self, dataloader: DataLoader, encoder, pooler) -> EmbedderResult:
        emb = compute_embeddings(dataloader, encoder, pooler,
                                 normalize=self.config.normalize_embeddings)
        ds = dataloader.dataset
        return EmbedderResult(embeddings=emb, text=list(ds.data),
                              metadata=ds.metadata)


def build_chunks(distances: np.ndarray,
                 breakpoint_percentile_threshold: int) -> list[tuple[int, int]]:
    """Percentile breakpoints -> [start, end) sentence index groups
    (reference semantic_chunk.py:58-102 semantics, incl. the
    one-group fallback for empty distances)."""
    if len(distances) == 0:
        return [(0, 0)]
    thr = np.percentile(distances, breakpoint_percentile_threshold)
    above = [i for i, x in enumerate(distances) if x > thr]
    groups = []
    start = 0
    for i in above:
        groups.append((start, i + 1))
        start = i + 1
    groups.append((start, len(distances) + 1))
    return groups


class SemanticChunkEmbedderConfig(BaseConfig):
    name: Literal["semantic_chunk"] = "semantic_chunk"
    breakpoint_percentile_threshold: int = Field(90)
    chunk_batch_size: int = Field(8)
    min_chunk_length: int = Field(750)
    normalize_embeddings: bool = Field(False)


class SemanticChunkEmbedder:
    """Two-pass semantic chunking (reference semantic_chunk.py:105-294):
    embed sentence buffers, break documents at cosine-distance
    percentile outliers, regroup sentences, re-embed the chunks."""

    def __init__(self, config: SemanticChunkEmbedderConfig):
        self.config = config

    def embed(self, dataloader: DataLoader, encoder, pooler) -> EmbedderResult:
        ds = dataloader.dataset
        if ds.metadata is None:
            raise ValueError("Metadata is required for semantic chunking.")
        if len(ds) == 0:
            # every buffer was filtered out (all sentences shorter than
            # the dataset's min_buffer_length): an empty shard, not an
            # error — one short file must not kill a distributed run.
            return EmbedderResult(
                embeddings=np.zeros((0, 0), dtype=np.float32),
                text=[], metadata=[])
        if ds.metadata[0].get("path") is None:
            raise ValueError("Metadata path is required for semantic "
                             "chunking.")

        # group buffers by document
        doc_bounds = []
        cur = 0
        cur_doc = ds.metadata[0]["path"]
        for i, md in enumerate(ds.metadata):
            if md["path"] != cur_doc:
                doc_bounds.append((cur, i))
                cur, cur_doc = i, md["path"]
        doc_bounds.append((cur, len(ds)))

        buffer_embeds = compute_embeddings(dataloader, encoder, pooler)

        # GPU adjacent-cosine distances per document
        dev = encoder.device
        emb_t = torch.from_numpy(buffer_embeds)
        if str(dev).startswith("cuda"):
            emb_t = emb_t.to(dev)
        dataset_indices: list[tuple[int, int]] = []
        for start, end in doc_bounds:
            seg = emb_t[start:end]
            if seg.shape[0] >= 2:
                distances = ops.adjacent_cosine(seg).cpu().numpy()
            else:
                distances = np.zeros(0, dtype=np.float32)
            for s, e in build_chunks(
                distances, self.config.breakpoint_percentile_threshold
            ):
                dataset_indices.append((start + s, start + e))

        data = []
        for s, e in dataset_indices:
            group = ds.metadata[s:e]
            data.append("".join(g["sentence"] for g in group))
        metadata = [dict(ds.metadata[s]) for s, _ in dataset_indices]

        keep = [i for i, x in enumerate(data)
                if len(x) > self.config.min_chunk_length]
        data = [data[i] for i in keep]
        metadata = [metadata[i] for i in keep]
        for md in metadata:
            md.pop("sentence", None)

        chunk_ds = InMemoryDataset(data, metadata)
        chunk_loader = DataLoader(
            pin_memory=dataloader.pin_memory,
            batch_size=self.config.chunk_batch_size,
            num_workers=dataloader.num_workers,
            dataset=chunk_ds,
            collate_fn=DataCollator(encoder.tokenizer),
        )
        chunk_embeds = compute_embeddings(
            chunk_loader, encoder, pooler,
            normalize=self.config.normalize_embeddings,
        )
        return EmbedderResult(embeddings=chunk_embeds, text=data,
                              metadata=metadata)


EmbedderConfigs = Union[FullSequenceEmbedderConfig,
                        SemanticChunkEmbedderConfig]

STRATEGIES: dict[str, tuple[type, type]] = {
    "full_sequence": (FullSequenceEmbedderConfig, FullSequenceEmbedder),
    "semantic_chunk": (SemanticChunkEmbedderConfig, SemanticChunkEmbedder),
}


def get_embedder(kwargs: dict[str, Any]):
    name = kwargs.get("name", "")
    if name not in STRATEGIES:
        raise ValueError(
            f"unknown embedder name {name!r}; options: {sorted(STRATEGIES)}"
        )
    cfg_cls, impl_cls = STRATEGIES[name]
    return impl_cls(cfg_cls(**kwargs))
