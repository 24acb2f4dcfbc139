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
    # defaults beside the dataset — reference FaissIndexV1 YAMLs
    # (rag/search.py V1, examples/evaluate/*.yaml) give only
    # dataset_dir
    faiss_index_path: Path = Field(
        Path(""),
        description="index cache path (packed codes + meta, .npz); "
        "empty -> <dataset_dir>/hip_index.npz",
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
    nprobe: int = 32  # ivf: chunks probed per query (recall@10 0.94
    # at 10M rows/nprobe 32, 0.97 at 64 — profiles/RESULTS.md)

    def model_post_init(self, _ctx) -> None:
        if str(self.faiss_index_path) in ("", "."):
            self.faiss_index_path = (Path(self.dataset_dir)
                                     / "hip_index.npz")


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

    def _assign(self, cent: torch.Tensor) -> torch.Tensor:
        """argmax cluster per corpus row, GEMM-chunked so the score
        matrix never exceeds ~1 GB at 10M+ rows."""
        rows = max(1, (1 << 28) // max(cent.shape[0], 1))
        outs = []
        for r0 in range(0, self.n, rows):
            outs.append(
                (self.corpus[r0:r0 + rows] @ cent.t()).argmax(1))
        return torch.cat(outs)

    def _build_ivf(self) -> None:
        """Batched IVF: GPU k-means, then clusters split into chunks of
        at most `cap` rows and stored as PADDED fp32 slabs
        [nchunks, cap, dim_pad4] with per-chunk row->corpus-id maps. Search
        scores every probed chunk with one regular kernel launch (no
        per-query host loop) and top-ks over the padded candidate
        matrix. Replaces the reference's faiss HNSW approximate option
        (rag/search.py:237,245) with a design sized for 10M+ corpora
        resident in 288 GB HBM."""
        C = self.nlist or max(int(self.n ** 0.5), 1)
        C = min(C, self.n)
        dev = self.corpus.device
        g = torch.Generator().manual_seed(0)
        init = torch.randperm(self.n, generator=g)[:C]
        cent = self.corpus[init.to(dev)].clone()
        ar = torch.arange(C, device=dev)
        for _ in range(10):
            assign = self._assign(cent)
            sums = torch.zeros_like(cent).index_add_(0, assign, self.corpus)
            cnt = torch.bincount(assign, minlength=C).float()
            keep = cnt > 0
            cent[keep] = sums[keep] / cnt[keep].unsqueeze(1)
        assign = self._assign(cent)
        order = torch.argsort(assign.cpu(), stable=True)
        counts = torch.bincount(assign.cpu(), minlength=C)
        offs = torch.zeros(C + 1, dtype=torch.long)
        offs[1:] = counts.cumsum(0)

        # split clusters into <=cap-row chunks; pad each to cap
        cap = max(64, (self.n + C - 1) // C)
        chunk_src: list[tuple[int, int]] = []  # (start, len) into order
        for c in range(C):
            o0, o1 = int(offs[c]), int(offs[c + 1])
            for s in range(o0, o1, cap):
                chunk_src.append((s, min(cap, o1 - s)))
        nch = len(chunk_src)
        idx_pad = torch.full((nch, cap), -1, dtype=torch.int32)
        for i, (s, ln) in enumerate(chunk_src):
            idx_pad[i, :ln] = order[s:s + ln].int()
        lens = torch.tensor([ln for _, ln in chunk_src],
                            dtype=torch.int32)
        # padded fp32 slabs: exact scores (bf16 slabs measured to
        # scramble near-tie ordering on tight clusters); 2x traffic is
        # still far below the exact path at small query counts. The dim
        # is padded to a multiple of 4 so the kernel's 16-byte row
        # loads stay naturally aligned for any embedding size.
        Hp = (self.dim + 3) // 4 * 4
        slabs = torch.zeros(nch, cap, Hp, dtype=torch.float32,
                            device=dev)
        gather = idx_pad.clamp(min=0).long().to(dev)
        slabs[:, :, :self.dim].copy_(
            self.corpus[gather.reshape(-1)].reshape(nch, cap, self.dim))
        # zero the pad rows (the clamp pointed them at corpus[0]): they
        # must not contribute to chunk means nor produce candidates
        valid = (torch.arange(cap, device=dev)[None, :]
                 < lens.to(dev)[:, None])
        slabs.mul_(valid.unsqueeze(-1))
        # per-chunk centroid = chunk mean (valid rows only)
        csum = slabs[:, :, :self.dim].sum(1)
        ccent = csum / lens.to(dev).float().clamp(min=1).unsqueeze(1)

        self.ivf_nlist = C
        self.ivf_cap = cap
        self.ivf_nchunks = nch
        self.ivf_centroids = ccent        # [nchunks, dim] chunk means
        self.ivf_slabs = slabs            # [nchunks, cap, dim_pad4] f32
        self.ivf_idx = idx_pad.to(dev)    # [nchunks, cap] int32, -1 pad
        self.ivf_len = lens.to(dev)       # [nchunks] int32

    def _topk_ivf(self, q: torch.Tensor, k: int):
        """Batched IVF search: score ALL (query, probe) chunks in one
        kernel launch (GPU: ivf_score HIP kernel; CPU: one bmm), then a
        single padded top-k. No per-query host loop."""
        Q = q.shape[0]
        nprobe = min(self.nprobe, self.ivf_nchunks)
        cscores = q @ self.ivf_centroids.t()
        _, probes = torch.topk(cscores, nprobe, dim=1)   # [Q, nprobe]
        probes = probes.int().contiguous()
        cap = self.ivf_cap
        Hp = self.ivf_slabs.shape[2]
        qp = (q if Hp == q.shape[1]
              else torch.nn.functional.pad(q, (0, Hp - q.shape[1])))
        if q.is_cuda and ops.HAVE_HIP:
            scores = ops.ivf_score(self.ivf_slabs, self.ivf_len,
                                   probes, qp)           # [Q, np, cap]
        else:
            sl = self.ivf_slabs[probes.long()]           # [Q, np, cap, D]
            scores = torch.einsum("qpcd,qd->qpc", sl, qp)
            valid = (torch.arange(cap, device=q.device)[None, None, :]
                     < self.ivf_len[probes.long()][:, :, None])
            scores = scores.masked_fill(~valid, float("-inf"))
        cand = self.ivf_idx[probes.long()].reshape(Q, -1)  # [Q, np*cap]
        flat = scores.reshape(Q, -1).contiguous()
        kk = min(k, flat.shape[1])
        v, i = ops.topk(flat, kk)
        idx = cand.gather(1, i.long())
        if kk < k:
            v = torch.nn.functional.pad(v, (0, k - kk),
                                        value=float("-inf"))
            idx = torch.nn.functional.pad(idx, (0, k - kk), value=-1)
        return v, idx

    # ---- search ---------------------------------------------------------
    def transform(self, embeddings: np.ndarray) -> np.ndarray:
        return normalize_L2(embeddings)

    def _topk_exact(self, q: torch.Tensor, k: int):
        """Chunked GEMM + HIP top-k over the fp32 corpus.

        The scoring GEMM stays a library call by policy: scores are
        fp32 (faiss-parity exactness) and fp32 GEMM peaks at 157 TF on
        gfx950 (the f32-input MFMA rate) — rocBLAS's tuned SGEMM
        (~147 TF) is already at that roofline, so a hand-written f32
        kernel has nothing to win here. The hand-written kernels own
        the top-k selection (radix-select, topk_search.hip), the
        Hamming path and the IVF chunk scorer."""
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
