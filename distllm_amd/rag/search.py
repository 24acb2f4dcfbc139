"""Semantic search: GPU-resident flat index + retriever, replacing
faiss-gpu entirely (reference distllm/rag/search.py:59-928; SURVEY.md
K11/K12/K14).

Design (MI355X-first): the corpus embedding matrix lives in HBM
(288 GB/GPU holds ~17B fp32x4 dims); exact inner-product search is a
chunked GEMM (hipBLASLt) + the hand-written radix-select top-k kernel
(topk_search.hip); the `ubinary` precision path packs sign bits into
u64 words (word-major for coalesced candidate reads) and scores with
the HIP popcount-Hamming kernel, then rescores `top_k *
rescore_multiplier` candidates against the fp32 queries.

Config-surface parity: `faiss_index_v2` is accepted as an alias of
`hip_index`; search_algorithm 'hnsw' falls back to exact (exact flat
search on GPU covers the reference's default configs; approximate
search is unnecessary below ~100M chunks at these speeds).
"""

from __future__ import annotations

import json
import time
from pathlib import Path
from typing import Any, NamedTuple

import numpy as np
import torch
from pydantic import Field

from .. import ops
from ..embed.encoders import get_encoder
from ..embed.poolers import get_pooler
from ..utils import BaseConfig, batch_data


class BatchedSearchResults(NamedTuple):
    total_scores: list[list[float]]
    total_indices: list[list[int]]


def normalize_L2(embeddings: np.ndarray) -> np.ndarray:
    """In-place row L2 normalize (faiss.normalize_L2 semantics)."""
    norms = np.linalg.norm(embeddings, axis=1, keepdims=True)
    np.divide(embeddings, np.maximum(norms, 1e-12), out=embeddings)
    return embeddings


def quantize_embeddings(emb: np.ndarray, precision: str) -> np.ndarray:
    """sentence-transformers-style quantization: 'ubinary' packs the
    sign bits (emb > 0) into uint8 rows."""
    if precision == "float32":
        return emb.astype(np.float32)
    if precision == "ubinary":
        return np.packbits(emb > 0, axis=1)
    raise ValueError(f"unsupported precision {precision}")


class HipIndexConfig(BaseConfig):
    name: str = Field("hip_index",
                      description="'hip_index' (alias: 'faiss_index_v2')")
    dataset_dir: Path
    faiss_index_path: Path = Field(
        description="index cache path (packed codes + meta, .npz)"
    )
    dataset_chunk_paths: list[Path] | None = None
    precision: str = "float32"
    search_algorithm: str = "exact"
    # The reference default is 2 (sentence-transformers semantics) but
    # sign-bit quantization recall is strongly multiplier-dependent
    # (recall@10 ~0.2 at x2 on clustered data, profiles/RESULTS.md);
    # default raised to 8 and a recall warning is emitted for ubinary.
    rescore_multiplier: int = 8
    num_quantization_workers: int = 1
    nlist: int = 0    # ivf: clusters (0 -> ~sqrt(N))
    nprobe: int = 8   # ivf: clusters searched per query


class HipFlatIndex:
    """GPU-resident flat index over the HF-dataset interchange format
    (columns `text`, `embeddings` fp32)."""

    def __init__(self, dataset_dir: Path, faiss_index_path: Path,
                 dataset_chunk_paths: list[Path] | None = None,
                 precision: str = "float32",
                 search_algorithm: str = "exact",
                 rescore_multiplier: int = 8,
                 num_quantization_workers: int = 1,
                 device: str | None = None,
                 score_chunk_rows: int = 1 << 21,
                 nlist: int = 0, nprobe: int = 8):
        if precision not in ("float32", "ubinary"):
            raise ValueError(f"invalid precision {precision}")
        if precision == "ubinary":
            import warnings

            warnings.warn(
                "[hip_index] ubinary (packed sign-bit) precision trades "
                "recall for a 32x smaller index; recall@k depends "
                "strongly on the data and on rescore_multiplier "
                f"(currently {rescore_multiplier}). Measure recall "
                "against precision='float32' on a sample before "
                "production use.", UserWarning, stacklevel=2)
        if search_algorithm not in ("exact", "hnsw", "ivf"):
            raise ValueError(f"invalid search_algorithm {search_algorithm}")
        if search_algorithm == "hnsw":
            print("[hip_index] 'hnsw' requested: exact GPU flat search is "
                  "used instead (covers reference defaults; 'ivf' gives "
                  "an approximate option)")
            search_algorithm = "exact"
        self.search_algorithm = search_algorithm
        self.nlist = nlist
        self.nprobe = nprobe
        self.precision = precision
        self.rescore_multiplier = rescore_multiplier
        self.faiss_index_path = Path(faiss_index_path)
        self.score_chunk_rows = score_chunk_rows
        self.device = device or (
            "cuda" if torch.cuda.is_available() else "cpu"
        )

        from datasets import load_from_disk

        self.dataset = load_from_disk(str(dataset_dir))

        if self.faiss_index_path.exists():
            print(f"Loading index from {self.faiss_index_path}")
            self._load()
        else:
            print(f"Creating index at {self.faiss_index_path}")
            self._create(dataset_chunk_paths)

    # ---- build / persist ------------------------------------------------
    def _embeddings_fp32(self, ds) -> np.ndarray:
        return np.asarray(ds["embeddings"], dtype=np.float32)

    def _create(self, chunk_paths: list[Path] | None) -> None:
        from datasets import load_from_disk

        mats = [self._embeddings_fp32(self.dataset)]
        if chunk_paths:
            for p in chunk_paths:
                mats.append(self._embeddings_fp32(load_from_disk(str(p))))
        emb = np.concatenate(mats, 0) if len(mats) > 1 else mats[0]
        self._install(emb)
        self.faiss_index_path.parent.mkdir(parents=True, exist_ok=True)
        payload = {"meta": json.dumps({"precision": self.precision,
                                       "dim": int(emb.shape[1])})}
        if self.precision == "ubinary":
            payload["codes"] = quantize_embeddings(emb, "ubinary")
            payload["embeddings"] = emb  # fp32 kept for rescoring
        else:
            payload["embeddings"] = emb
        np.savez(self.faiss_index_path, **payload)

    def _load(self) -> None:
        z = np.load(self.faiss_index_path, allow_pickle=False)
        emb = z["embeddings"].astype(np.float32)
        self._install(emb)

    def _install(self, emb: np.ndarray) -> None:
        dev = self.device
        self.dim = emb.shape[1]
        self.n = emb.shape[0]
        self.corpus = torch.from_numpy(np.ascontiguousarray(emb)).to(dev)
        if self.precision == "ubinary":
            codes = quantize_embeddings(emb, "ubinary")  # [N, dim/8] u8
            W = (codes.shape[1] + 7) // 8
            pad = W * 8 - codes.shape[1]
            if pad:
                codes = np.pad(codes, ((0, 0), (0, pad)))
            c64 = codes.reshape(self.n, W, 8).copy().view(np.uint64)
            c64 = c64.reshape(self.n, W).T.copy()  # word-major [W, N]
            self.codes_t = torch.from_numpy(c64.view(np.int64)).to(dev)
            self.code_words = W
        if self.search_algorithm == "ivf":
            self._build_ivf()

    def _build_ivf(self) -> None:
        """Simple GPU k-means (inner-product on normalized centroids);
        corpus reordered cluster-major for contiguous candidate gathers.
        Replaces the reference's faiss HNSW approximate option
        (rag/search.py:237,245) with an IVF-flat design that maps to
        GEMM + gather on this hardware."""
        C = self.nlist or max(int(self.n ** 0.5), 1)
        C = min(C, self.n)
        g = torch.Generator().manual_seed(0)
        init = torch.randperm(self.n, generator=g)[:C]
        cent = self.corpus[init.to(self.corpus.device)].clone()
        for _ in range(10):
            assign = (self.corpus @ cent.t()).argmax(1)
            for c in range(C):
                m = assign == c
                if m.any():
                    cent[c] = self.corpus[m].mean(0)
        assign = (self.corpus @ cent.t()).argmax(1)
        order = torch.argsort(assign, stable=True)
        counts = torch.bincount(assign, minlength=C)
        offs = torch.zeros(C + 1, dtype=torch.long)
        offs[1:] = counts.cumsum(0).cpu()
        self.ivf_centroids = cent
        self.ivf_order = order            # corpus row ids, cluster-major
        self.ivf_offsets = offs           # [C+1] boundaries into order
        self.ivf_nlist = C
        # cluster-major slab copy: probing then reads CONTIGUOUS rows
        # (a [len, D] @ [D] GEMV per probed cluster) instead of a random
        # row gather — 384x less traffic than gathering candidate rows.
        # Costs one corpus duplicate; by design for 288 GB HBM.
        self.corpus_sorted = self.corpus[order.to(self.corpus.device)]\
            .contiguous()

    def _topk_ivf(self, q: torch.Tensor, k: int):
        """IVF probe scoring over cluster-major contiguous slabs.

        IVF is the LOW-LATENCY single/small-query path: it reads only
        the probed clusters' rows. For large query batches the exact
        path is faster on this GPU — one GEMM reads the corpus once for
        ALL queries, while IVF re-reads per query (measured in
        profiles/RESULTS.md)."""
        nprobe = min(self.nprobe, self.ivf_nlist)
        cscores = q @ self.ivf_centroids.t()
        _, probes = torch.topk(cscores, nprobe, dim=1)   # [Q, nprobe]
        probes_h = probes.cpu().tolist()
        vs, is_ = [], []
        for qi in range(q.shape[0]):
            parts_v, parts_i = [], []
            for c in probes_h[qi]:
                o0 = int(self.ivf_offsets[c])
                o1 = int(self.ivf_offsets[c + 1])
                if o0 == o1:
                    continue
                parts_v.append(self.corpus_sorted[o0:o1] @ q[qi])
                parts_i.append(self.ivf_order[o0:o1])
            scores = torch.cat(parts_v).unsqueeze(0)
            cand = torch.cat(parts_i).to(scores.device)
            kk = min(k, cand.numel())
            v, i = ops.topk(scores.contiguous(), kk)
            idx = cand[i[0].long()].int()
            if kk < k:
                v = torch.nn.functional.pad(v, (0, k - kk),
                                            value=float("-inf"))
                idx = torch.nn.functional.pad(idx, (0, k - kk), value=-1)
            vs.append(v[0])
            is_.append(idx)
        return torch.stack(vs), torch.stack(is_)

    # ---- search ---------------------------------------------------------
    def transform(self, embeddings: np.ndarray) -> np.ndarray:
        return normalize_L2(embeddings)

    def _topk_exact(self, q: torch.Tensor, k: int):
        """Chunked GEMM + HIP top-k over the fp32 corpus."""
        N = self.n
        k = min(k, N)
        chunk = self.score_chunk_rows
        if N <= chunk:
            scores = q @ self.corpus.t()
            return ops.topk(scores.contiguous(), k)
        vs, is_ = [], []
        for c0 in range(0, N, chunk):
            c1 = min(c0 + chunk, N)
            scores = q @ self.corpus[c0:c1].t()
            v, i = ops.topk(scores.contiguous(), min(k, c1 - c0))
            vs.append(v)
            is_.append(i + c0)
        allv = torch.cat(vs, 1)
        alli = torch.cat(is_, 1)
        v, sel = torch.topk(allv, k, dim=1)
        return v, alli.gather(1, sel.long()).int()

    def _topk_ubinary(self, qfp: torch.Tensor, k: int):
        """Hamming candidates (HIP popcount) + fp32 rescore."""
        qb = quantize_embeddings(qfp.cpu().numpy(), "ubinary")
        W = self.code_words
        pad = W * 8 - qb.shape[1]
        if pad:
            qb = np.pad(qb, ((0, 0), (0, pad)))
        q64 = qb.reshape(qb.shape[0], W, 8).copy().view(np.uint64)
        q64 = torch.from_numpy(q64.reshape(qb.shape[0], W).view(np.int64))
        q64 = q64.to(self.corpus.device)
        scores = ops.hamming_score(q64, self.codes_t)  # [Q, N] = -dist
        kk = min(k * self.rescore_multiplier, self.n)
        _, cand = ops.topk(scores, kk)
        # rescore candidates with fp32 inner product
        gathered = self.corpus[cand.long()]            # [Q, kk, H]
        res = torch.einsum("qh,qkh->qk", qfp.to(self.corpus.device),
                           gathered)
        v, sel = torch.topk(res, min(k, kk), dim=1)
        return v, cand.gather(1, sel).int()

    def search(self, query_embedding: np.ndarray, top_k: int = 1,
               score_threshold: float = 0.0) -> BatchedSearchResults:
        t0 = time.perf_counter()
        q = torch.from_numpy(
            np.ascontiguousarray(query_embedding, dtype=np.float32)
        ).to(self.corpus.device)
        if self.precision == "ubinary":
            v, i = self._topk_ubinary(q, top_k)
        elif self.search_algorithm == "ivf":
            v, i = self._topk_ivf(q, top_k)
        else:
            v, i = self._topk_exact(q, top_k)
        v = v.cpu().tolist()
        i = i.cpu().tolist()
        print(f"Search time: {time.perf_counter() - t0:.6f} seconds")
        # IVF pads short candidate lists with index -1 / score -inf;
        # drop those unconditionally (not only under score_threshold) so
        # -1 never reaches get()/dataset indexing as "last row".
        vv, ii = [], []
        for scores, indices in zip(v, i):
            keep = [(s, ix) for s, ix in zip(scores, indices) if ix >= 0]
            vv.append([s for s, _ in keep])
            ii.append([ix for _, ix in keep])
        results = BatchedSearchResults(total_scores=vv, total_indices=ii)
        return self._filter_search_by_score(results, score_threshold)

    def _filter_search_by_score(self, results: BatchedSearchResults,
                                score_threshold: float
                                ) -> BatchedSearchResults:
        if not score_threshold:
            return results
        ti, ts = [], []
        for indices, scores in zip(results.total_indices,
                                   results.total_scores):
            ni = [ix for ix, s in zip(indices, scores)
                  if s >= score_threshold]
            ns = [s for s in scores if s >= score_threshold]
            ti.append(ni)
            ts.append(ns)
        return BatchedSearchResults(total_scores=ts, total_indices=ti)

    def get(self, indices: list[int], key: str) -> list[Any]:
        return [self.dataset[int(i)][key] for i in indices]


# alias names accepted in YAML
FaissIndexV2 = HipFlatIndex
FaissIndexV2Config = HipIndexConfig


class RetrieverConfig(BaseConfig):
    faiss_config: HipIndexConfig
    encoder_config: dict[str, Any]
    pooler_config: dict[str, Any]
    batch_size: int = 4

    def get_retriever(self) -> "Retriever":
        encoder = get_encoder(dict(self.encoder_config))
        pooler = get_pooler(dict(self.pooler_config))
        kwargs = self.faiss_config.model_dump(exclude={"name"})
        index = HipFlatIndex(**kwargs)
        return Retriever(encoder=encoder, pooler=pooler, faiss_index=index,
                         batch_size=self.batch_size)


class Retriever:
    """Query encoding + index search (reference rag/search.py:715-928)."""

    def __init__(self, encoder, pooler, faiss_index: HipFlatIndex,
                 batch_size: int = 4):
        self.encoder = encoder
        self.pooler = pooler
        self.faiss_index = faiss_index
        self.batch_size = batch_size

    def search(self, query: str | list[str] | None = None,
               query_embedding: np.ndarray | None = None, top_k: int = 1,
               score_threshold: float = 0.0
               ) -> tuple[BatchedSearchResults, np.ndarray]:
        if query is None and query_embedding is None:
            raise ValueError("Provide at least one of query or "
                             "query_embedding.")
        if query_embedding is None:
            query_embedding = self.get_pooled_embeddings(query)
        results = self.faiss_index.search(query_embedding, top_k,
                                          score_threshold)
        return results, query_embedding

    def get_pooled_embeddings(self, query: str | list[str]) -> np.ndarray:
        """Length-sorted batched query embedding with original-order
        restore (reference search.py:800-881)."""
        if isinstance(query, str):
            query = [query]
        order = sorted(range(len(query)), key=lambda i: len(query[i]))
        sorted_q = [query[i] for i in order]
        embs = []
        for batch in batch_data(sorted_q, self.batch_size):
            embs.append(self._embed_batch(batch))
        emb = np.concatenate(embs, 0)
        out = np.empty_like(emb)
        out[np.asarray(order)] = emb
        return self.faiss_index.transform(out)

    @torch.no_grad()
    def _embed_batch(self, batch: list[str]) -> np.ndarray:
        enc = self.encoder.tokenizer(batch, padding=True, truncation=True,
                                     return_tensors="pt",
                                     pad_to_multiple_of=8)
        hidden = self.encoder.encode(enc)
        mask = enc["attention_mask"].to(hidden.device)
        pooled = self.pooler.pool(hidden, mask)
        return pooled.float().cpu().numpy()

    def get(self, indices: list[int], key: str) -> list[Any]:
        return self.faiss_index.get(indices, key)

    def get_embeddings(self, indices: list[int]) -> np.ndarray:
        return np.asarray(self.faiss_index.get(indices, "embeddings"),
                          dtype=np.float32)

    def get_texts(self, indices: list[int]) -> list[str]:
        return self.faiss_index.get(indices, "text")
