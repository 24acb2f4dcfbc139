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
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch-size", type=int, default=512)
    ap.add_argument("--seq-len", type=int, default=512)
    ap.add_argument("--model", type=str, default="pubmedbert-base")
    ap.add_argument("--mode", type=str, default="embed",
                    choices=["embed", "generate"],
                    help="embed = flagship seqs/s; generate = decode "
                         "tokens/s (Llama-class, paged KV)")
    ap.add_argument("--pooler", type=str, default="mean",
                    choices=["mean", "last_token"])
    ap.add_argument("--prompt-len", type=int, default=128)
    args = ap.parse_args()

    if args.mode == "generate":
        bench_generate(args)
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
                       max_model_len=P + args.steps + args.warmup + 8)
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
            "dtype": "bf16" if have_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": model_name,
                "global_batch": world * B,
                "seq_len": P,
                "parallelism": f"dp{world}",
            },
        }))


if __name__ == "__main__":
    main()
