"""Worker script for tensor-parallel correctness tests (gloo, world=2).
Run under torchrun by tests/test_distributed.py."""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

import torch
import torch.distributed as dist

from distllm_amd.engine import LLMEngine, SamplingParams
from distllm_amd.models import build_model, get_dims
from distllm_amd.models.loader import load_weights
from distllm_amd.models.mistral import DecoderModel


def hf_style_state(model):
    """Dump a full (non-TP) native model as HF-Llama-named tensors."""
    sd = {}
    sd["model.embed_tokens.weight"] = model.tok_emb
    sd["model.norm.weight"] = model.final_norm
    if model.lm_head is not None:
        sd["lm_head.weight"] = model.lm_head
    for i, lyr in enumerate(model.layers):
        p = f"model.layers.{i}."
        sd[p + "self_attn.q_proj.weight"] = lyr.wq
        sd[p + "self_attn.k_proj.weight"] = lyr.wk
        sd[p + "self_attn.v_proj.weight"] = lyr.wv
        sd[p + "self_attn.o_proj.weight"] = lyr.wo
        sd[p + "input_layernorm.weight"] = lyr.in_norm
        sd[p + "post_attention_layernorm.weight"] = lyr.post_norm
        sd[p + "mlp.gate_proj.weight"] = lyr.wgate
        sd[p + "mlp.up_proj.weight"] = lyr.wup
        sd[p + "mlp.down_proj.weight"] = lyr.wdown
    return {k: v.detach().clone().contiguous() for k, v in sd.items()}


def main():
    tmp = Path(sys.argv[1])
    dist.init_process_group("gloo")
    rank = dist.get_rank()

    torch.manual_seed(123)
    ref = build_model("llama-tiny", dtype=torch.float32)
    if rank == 0:
        from safetensors.torch import save_file

        save_file(hf_style_state(ref), str(tmp / "model.safetensors"))
    dist.barrier()

    dims = get_dims("llama-tiny")
    tp_model = DecoderModel(dims, dtype=torch.float32,
                            tp_group=dist.group.WORLD)
    assert tp_model.tp == 2
    load_weights(tp_model, tmp)
    tp_model.eval()

    ids = torch.tensor([[5, 9, 100, 42, 7, 3]])
    mask = torch.ones_like(ids, dtype=torch.int32)
    with torch.no_grad():
        h_ref = ref(ids, mask)
        h_tp = tp_model(ids, mask)
    err = (h_ref - h_tp).abs().max().item()
    assert err < 1e-4, f"TP forward mismatch: {err}"

    # engine decode under TP matches single-rank greedy
    engine = LLMEngine(tp_model, max_batch_size=2, max_model_len=64,
                       num_blocks=32)
    params = SamplingParams(temperature=0.0, min_p=0.0, max_tokens=5)
    out = engine.generate([[5, 9, 100, 42]], params)[0]

    ref_ids = [5, 9, 100, 42]
    want = []
    for _ in range(5):
        t = torch.tensor([ref_ids])
        m = torch.ones_like(t, dtype=torch.int32)
        with torch.no_grad():
            logits = ref.logits(ref(t, m)[:, -1])
        nxt = int(logits.argmax(-1))
        want.append(nxt)
        ref_ids.append(nxt)
    assert out == want, f"TP decode {out} != ref {want}"
    if rank == 0:
        print("TP_OK")


if __name__ == "__main__":
    main()
