"""fp8-e4m3 KV-cache decode throughput vs bf16 KV at long context.

The paged-attention kernel is templated on cache type; at long context
the decode step is KV-bandwidth-bound, so fp8 halves the bytes read.

    python profiles/fp8_kv_probe.py
"""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import torch

from distllm_amd.engine import LLMEngine, SamplingParams
from distllm_amd.engine.engine import _Seq
from distllm_amd.models import build_model


def bench(kv_dtype, B=128, P=2048, steps=20):
    torch.manual_seed(0)
    model = build_model("llama3-8b", device="cuda:0", dtype=torch.bfloat16)
    engine = LLMEngine(model, max_batch_size=B, max_model_len=P + steps + 8,
                       kv_cache_dtype=kv_dtype)
    params = SamplingParams(temperature=0.5, min_p=0.1,
                            max_tokens=steps + 4, stop_token_ids=())
    prompts = torch.randint(4, model.dims.vocab_size, (B, P)).tolist()
    seqs = [_Seq(seq_id=i, prompt_ids=p) for i, p in enumerate(prompts)]
    for s in seqs:
        s.blocks = engine.blockman.alloc(
            engine.blockman.blocks_for(P + params.max_tokens + 1))
    res = {}
    engine._run_prefill(seqs, params, res)
    for _ in range(3):
        engine._run_decode(seqs, params, res)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        engine._run_decode(seqs, params, res)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    print(f"kv={kv_dtype}: {B / dt / 1000:.1f}k tok/s decode "
          f"({dt * 1e3:.2f} ms/step, ctx {P})")
    del engine, model
    torch.cuda.empty_cache()


if __name__ == "__main__":
    bench("bf16")
    bench("fp8_e4m3")
