"""Worker for the tp=2 on-GPU test (RCCL world=2, one rank per GPU).
Run under torchrun by tests/test_tp_gpu.py when >=2 GPUs are visible.

Verifies: (1) tp=2 greedy decode THROUGH THE ENGINE (hipGraph capture
with the per-layer RCCL all-reduces IN-GRAPH) produces
the same tokens as a single-GPU tp=1 engine with the same weights;
(2) both ranks agree.
"""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import torch
import torch.distributed as dist

from distllm_amd.engine import LLMEngine, SamplingParams
from distllm_amd.models import build_model, get_dims
from distllm_amd.models.loader import load_weights
from distllm_amd.models.mistral import DecoderModel
from tp_worker import hf_style_state


def main():
    tmp = Path(sys.argv[1])
    dist.init_process_group("nccl")
    rank = dist.get_rank()
    torch.cuda.set_device(rank)
    dev = f"cuda:{rank}"

    torch.manual_seed(123)
    ref = build_model("llama-tiny", dtype=torch.bfloat16)
    if rank == 0:
        from safetensors.torch import save_file

        save_file(hf_style_state(ref), str(tmp / "model.safetensors"))
    dist.barrier()

    dims = get_dims("llama-tiny")
    prompt = [5, 9, 100, 42, 7, 3]
    params = SamplingParams(temperature=0.0, min_p=0.0, max_tokens=12,
                            stop_token_ids=())

    # tp=1 reference tokens (rank-local single-GPU engine, graphs on)
    ref1 = DecoderModel(dims, dtype=torch.bfloat16).to(dev)
    load_weights(ref1, tmp)
    ref1.eval()
    eng1 = LLMEngine(ref1, max_batch_size=2, max_model_len=64)
    out1 = eng1.generate([prompt], params)[0]

    # tp=2 engine: graphs capture the RCCL all-reduces in-graph
    tp_model = DecoderModel(dims, dtype=torch.bfloat16,
                            tp_group=dist.group.WORLD).to(dev)
    assert tp_model.tp == 2
    load_weights(tp_model, tmp)
    tp_model.eval()
    eng2 = LLMEngine(tp_model, max_batch_size=2, max_model_len=64)
    out2 = eng2.generate([prompt], params)[0]

    graphs_live = eng2.use_graphs and len(eng2._graphs) > 0
    assert out2 == out1, f"rank {rank}: tp2 {out2} != tp1 {out1}"

    # cross-rank agreement
    t = torch.tensor(out2, dtype=torch.long, device=dev)
    tt = [torch.zeros_like(t) for _ in range(2)]
    dist.all_gather(tt, t)
    assert torch.equal(tt[0], tt[1]), "ranks diverged"

    if rank == 0:
        print(f"TP2_GPU_OK graphs_in_use={graphs_live} tokens={out2}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
