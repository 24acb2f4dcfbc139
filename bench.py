"""Flagship benchmark: PubMedBERT bs512 embedding throughput (seqs/sec).

Driver contract: `python bench.py --gpus N --steps K --warmup W`.
For N>1 the driver launches this under torch.distributed.run with one
rank per GPU (RCCL over xGMI). One timed step = embed one batch of
`--batch-size` synthetic sequences of length `--seq-len` through the
native HIP-kernel encoder forward + fused mean-pool + L2 normalize +
async D2H into a pinned host buffer — the hot loop of the reference's
headline config (examples/embed/AMP.nougat.pubmedbert.yaml:
batch_size=512, maxlen 512, mean pool).

Rank 0 prints ONE JSON line with the whole-job aggregate.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch

# Synthetic benchmark: the hash tokenizer is the sanctioned stand-in.
os.environ.setdefault("DISTLLM_ALLOW_HASH_TOKENIZER", "1")


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch-size", type=int, default=512)
    ap.add_argument("--seq-len", type=int, default=512)
    ap.add_argument("--model", type=str, default="pubmedbert-base")
    ap.add_argument("--mode", type=str, default="embed",
                    choices=["embed", "generate", "pipeline"],
                    help="embed = flagship seqs/s; generate = decode "
                         "tokens/s (Llama-class, paged KV); pipeline = "
                         "full jsonl_chunk -> semantic_chunk embed -> "
                         "write (tokenizer + staging included)")
    ap.add_argument("--docs-per-file", type=int, default=128)
    ap.add_argument("--pooler", type=str, default="mean",
                    choices=["mean", "last_token"])
    ap.add_argument("--prompt-len", type=int, default=128)
    ap.add_argument("--weight-dtype", type=str, default="bf16",
                    choices=["bf16", "fp8_e4m3"],
                    help="generate mode: W8A16 fp8 decode weights")
    args = ap.parse_args()

    if args.mode == "generate":
        bench_generate(args)
        return
    if args.mode == "pipeline":
        bench_pipeline(args)
        return

    from distllm_amd.models import build_model
    from distllm_amd.parallel import (barrier, get_rank, get_world_size,
                                      init_distributed)
    from distllm_amd import ops

    rank, world = init_distributed()
    have_gpu = torch.cuda.is_available()
    if have_gpu:
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(local_rank)
        device = f"cuda:{local_rank}"
        dtype = torch.bfloat16
    else:
        device, dtype = "cpu", torch.float32

    torch.manual_seed(1234 + rank)
    model = build_model(args.model, device=device, dtype=dtype)
    dims = model.dims

    B, S = args.batch_size, args.seq_len
    ids = torch.randint(4, dims.vocab_size, (B, S), device=device)
    mask = torch.ones(B, S, device=device, dtype=torch.int32)
    host = torch.empty(B, dims.hidden_size, dtype=torch.float32,
                       pin_memory=have_gpu)
    # pinned-host staged inputs: each step re-uploads (H2D in the loop,
    # like the real dataloader path)
    ids_host = ids.to("cpu").pin_memory() if have_gpu else ids
    mask_host = mask.to("cpu").pin_memory() if have_gpu else mask

    @torch.no_grad()
    def step() -> None:
        d_ids = ids_host.to(device, non_blocking=True)
        d_mask = mask_host.to(device, non_blocking=True)
        hidden = model(d_ids, d_mask)
        if args.pooler == "last_token":
            pooled = ops.last_token_pool(hidden, d_mask, normalize=True)
        else:
            pooled = ops.mean_pool(hidden, d_mask, normalize=True)
        host.copy_(pooled, non_blocking=True)

    for _ in range(args.warmup):
        step()
    barrier()
    if have_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if have_gpu:
        torch.cuda.synchronize()
    barrier()
    elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks -> whole-job throughput
    if world > 1:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if have_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    value = world * B * args.steps / elapsed

    metric = ("seqs/sec embedded (PubMedBERT bs512)"
              if args.model == "pubmedbert-base"
              else f"seqs/sec embedded ({args.model} bs{B})")
    if rank == 0:
        print(json.dumps({
            "metric": metric,
            "value": round(value, 2),
            "unit": "seqs/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if have_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": world * B,
                "seq_len": S,
                "parallelism": f"dp{world}",
            },
        }))


def bench_generate(args) -> None:
    """Decode throughput: one step = one continuous-batching decode step
    for the whole batch (paged-KV attention + sampling)."""
    import torch

    from distllm_amd import ops
    from distllm_amd.engine import LLMEngine, SamplingParams
    from distllm_amd.models import build_model
    from distllm_amd.parallel import barrier, init_distributed

    rank, world = init_distributed()
    have_gpu = torch.cuda.is_available()
    if have_gpu:
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(local_rank)
        device, dtype = f"cuda:{local_rank}", torch.bfloat16
    else:
        device, dtype = "cpu", torch.float32

    model_name = args.model if args.model != "pubmedbert-base" \
        else "llama3-8b"
    torch.manual_seed(99 + rank)
    model = build_model(model_name, device=device, dtype=dtype)
    B, P = args.batch_size, args.prompt_len
    engine = LLMEngine(model, max_batch_size=B,
                       max_model_len=P + args.steps + args.warmup + 8,
                       weight_dtype=args.weight_dtype)
    params = SamplingParams(temperature=0.5, min_p=0.1,
                            max_tokens=args.steps + args.warmup + 4,
                            stop_token_ids=())

    # set up B running sequences via one prefill
    prompts = torch.randint(4, model.dims.vocab_size, (B, P)).tolist()
    from distllm_amd.engine.engine import _Seq

    seqs = [_Seq(seq_id=i, prompt_ids=p) for i, p in enumerate(prompts)]
    for s in seqs:
        s.blocks = engine.blockman.alloc(
            engine.blockman.blocks_for(P + params.max_tokens + 1))
    results: dict = {}
    engine._run_prefill(seqs, params, results)

    def step():
        engine._run_decode(seqs, params, results)

    for _ in range(args.warmup):
        step()
    barrier()
    if have_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if have_gpu:
        torch.cuda.synchronize()
    barrier()
    elapsed = time.perf_counter() - t0
    if world > 1:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if have_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    value = world * B * args.steps / elapsed
    if rank == 0:
        print(json.dumps({
            "metric": f"generate tokens/sec ({model_name} decode)",
            "value": round(value, 2),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": ("bf16" if args.weight_dtype == "bf16"
                      else "bf16+fp8w") if have_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": model_name,
                "global_batch": world * B,
                "seq_len": P,
                "parallelism": f"dp{world}",
            },
        }))


def bench_pipeline(args) -> None:
    """End-to-end semantic-chunk pipeline throughput:
    one step = one synthetic jsonl file through
    tokenize/split/buffer -> embed buffers -> GPU distance ->
    percentile chunking -> re-embed chunks -> write shard — the
    BASELINE config #2 wording ("semantic_chunk embedder") measured as
    a whole, host staging included. Reports docs/s and seqs/s plus a
    tokenize-vs-GPU breakdown. Single-rank (the DP story is covered by
    --mode embed)."""
    import random
    import shutil
    import tempfile
    from pathlib import Path

    from distllm_amd.distributed_embedding import embedding_worker

    have_gpu = torch.cuda.is_available()
    torch.manual_seed(1234)
    rng = random.Random(7)
    words = [f"w{i:03d}" for i in range(400)] + [
        "protein", "cell", "model", "study", "effect", "binding",
        "analysis", "method", "result", "treatment",
    ]

    def synth_doc() -> str:
        sents = []
        for _ in range(rng.randint(25, 60)):
            n = rng.randint(12, 24)
            s = " ".join(rng.choice(words) for _ in range(n))
            sents.append(s.capitalize() + ".")
        return " ".join(sents)

    root = Path(tempfile.mkdtemp(prefix="bench_pipeline_"))
    nfiles = args.warmup + args.steps
    files = []
    for f in range(nfiles):
        p = root / f"part_{f:03d}.jsonl"
        with open(p, "w") as fh:
            for _ in range(args.docs_per_file):
                fh.write(json.dumps({"text": synth_doc(), "path": f"{p.name}#{_}"}) + "\n")
        files.append(p)

    dataset_kwargs = {"name": "jsonl_chunk", "text_field": "text",
                      "batch_size": args.batch_size,
                      "num_data_workers": 4, "buffer_size": 4}
    encoder_kwargs = {"name": "auto",
                      "pretrained_model_name_or_path": args.model}
    pooler_kwargs = {"name": "mean"}
    embedder_kwargs = {"name": "semantic_chunk",
                       "chunk_batch_size": args.batch_size,
                       "normalize_embeddings": True}
    writer_kwargs = {"name": "numpy"}

    out = root / "out"
    out.mkdir()

    n_docs = n_bufs = n_chunks = 0
    t_total = 0.0
    for i, p in enumerate(files):
        timed = i >= args.warmup
        if have_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        embedding_worker(p, out, dataset_kwargs, encoder_kwargs,
                         pooler_kwargs, embedder_kwargs, writer_kwargs)
        if have_gpu:
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        if timed:
            t_total += dt
            n_docs += args.docs_per_file

    # breakdown on one extra file: host (tokenize/split/buffer) vs rest
    from distllm_amd.embed import (get_dataset, get_embedder, get_encoder,
                                   get_pooler)

    enc = get_encoder(encoder_kwargs, register=True)
    ds = get_dataset(dataset_kwargs)
    t0 = time.perf_counter()
    dl = ds.get_dataloader(files[0], enc)
    t_host = time.perf_counter() - t0
    n_bufs = len(dl.dataset)
    emb = get_embedder(embedder_kwargs)
    t0 = time.perf_counter()
    res = emb.embed(dl, enc, get_pooler(pooler_kwargs))
    if have_gpu:
        torch.cuda.synchronize()
    t_embed = time.perf_counter() - t0
    n_chunks = len(res.text)

    value = n_docs / t_total
    print(json.dumps({
        "metric": f"docs/sec semantic-chunk embedded ({args.model})",
        "value": round(value, 2),
        "unit": "docs/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(t_total / args.steps * 1000.0, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16" if have_gpu else "fp32",
        "data": "synthetic",
        "config": {
            "model": args.model,
            "docs_per_file": args.docs_per_file,
            "batch_size": args.batch_size,
            "embedder": "semantic_chunk",
            "parallelism": "dp1",
        },
        "breakdown": {
            "host_tokenize_split_s_per_file": round(t_host, 3),
            "embed_chunk_write_s_per_file": round(t_embed, 3),
            "buffers_per_file": n_bufs,
            "chunks_per_file": n_chunks,
            "buffers_per_s": round(n_bufs * args.steps / t_total, 1),
        },
    }))
    shutil.rmtree(root, ignore_errors=True)


if __name__ == "__main__":
    main()


