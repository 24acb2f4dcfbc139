"""Batched IVF vs exact flat search at 10M rows.

Synthetic clustered corpus (the regime ANN exists for): 10M x 384 fp32,
drawn around 20k centers. Measures recall@10 vs exact, batched QPS at
several batch sizes, and single-query latency.
"""

import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import torch

import distllm_amd.ops as ops  # noqa: E402
from distllm_amd.rag.search import HipFlatIndex  # noqa: E402


def make_index(corpus, nlist, nprobe):
    idx = object.__new__(HipFlatIndex)
    idx.device = corpus.device
    idx.corpus = corpus
    idx.n, idx.dim = corpus.shape
    idx.nlist, idx.nprobe = nlist, nprobe
    idx.ivf_nchunks = 0
    idx.score_chunk_rows = 1 << 21
    idx.precision = "float32"
    return idx


def bench(fn, iters=5):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    assert torch.cuda.is_available() and ops.HAVE_HIP
    dev = "cuda"
    N, H, NC = 10_000_000, 384, 20000
    g = torch.Generator(device=dev).manual_seed(0)
    centers = torch.randn(NC, H, device=dev, generator=g)
    corpus = torch.empty(N, H, device=dev)
    bs = 1_000_000
    for r0 in range(0, N, bs):
        n = min(bs, N - r0)
        ci = torch.randint(0, NC, (n,), device=dev, generator=g)
        corpus[r0:r0 + n] = centers[ci] + 0.35 * torch.randn(
            n, H, device=dev, generator=g)
    corpus = torch.nn.functional.normalize(corpus, dim=1)
    print(f"corpus: {N}x{H} fp32 = {corpus.numel() * 4 / 2**30:.1f} GiB")

    idx = make_index(corpus, nlist=4096, nprobe=64)
    t0 = time.perf_counter()
    idx._build_ivf()
    torch.cuda.synchronize()
    print(f"IVF build (kmeans {idx.ivf_nlist} lists -> "
          f"{idx.ivf_nchunks} chunks cap {idx.ivf_cap}): "
          f"{time.perf_counter() - t0:.1f}s")

    qn = 256
    queries = torch.nn.functional.normalize(
        corpus[torch.randint(0, N, (qn,), generator=g, device=dev)]
        + 0.1 * torch.randn(qn, H, device=dev, generator=g), dim=1)

    ve, ie = idx._topk_exact(queries, 10)
    for nprobe in (32, 64):
        idx.nprobe = nprobe
        vi, ii = idx._topk_ivf(queries, 10)
        rec = (ii.unsqueeze(2) == ie.unsqueeze(1)).any(-1).float() \
            .mean().item()
        print(f"nprobe={nprobe}: recall@10 vs exact = {rec:.3f}")

    for Q in (1, 16, 64, 256):
        qq = queries[:Q].contiguous()
        te = bench(lambda: idx._topk_exact(qq, 10))
        ti = bench(lambda: idx._topk_ivf(qq, 10))
        print(f"Q={Q:4d}: exact {te * 1e3:8.2f} ms ({Q / te:9.0f} q/s)  "
              f"ivf {ti * 1e3:8.2f} ms ({Q / ti:9.0f} q/s)  "
              f"speedup x{te / ti:.2f}")


if __name__ == "__main__":
    main()
