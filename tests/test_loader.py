"""Round-trip tests of the HF-checkpoint name mapping
(models/loader.py): export a native model's weights under HF names,
load them into a fresh random-init model, and require identical
forward outputs. Covers BertModel and Llama layouts plus the native
passthrough and the shape guard."""

import math

import pytest
import torch

from distllm_amd.models import build_model
from distllm_amd.models.loader import load_weights


def _save(tmp_path, state):
    from safetensors.torch import save_file

    tmp_path.mkdir(parents=True, exist_ok=True)
    save_file({k: v.contiguous() for k, v in state.items()},
              str(tmp_path / "model.safetensors"))


def _bert_to_hf(model) -> dict:
    p = dict(model.named_parameters())
    H = model.dims.hidden_size
    out = {
        "bert.embeddings.word_embeddings.weight": p["word_emb"],
        "bert.embeddings.position_embeddings.weight": p["pos_emb"],
        "bert.embeddings.token_type_embeddings.weight": p["type_emb"],
        "bert.embeddings.LayerNorm.weight": p["emb_ln_w"],
        "bert.embeddings.LayerNorm.bias": p["emb_ln_b"],
    }
    for i in range(len(model.layers)):
        lp = f"bert.encoder.layer.{i}."
        q, k, v = p[f"layers.{i}.wqkv"].split(H, dim=0)
        qb, kb, vb = p[f"layers.{i}.bqkv"].split(H, dim=0)
        out[lp + "attention.self.query.weight"] = q
        out[lp + "attention.self.key.weight"] = k
        out[lp + "attention.self.value.weight"] = v
        out[lp + "attention.self.query.bias"] = qb
        out[lp + "attention.self.key.bias"] = kb
        out[lp + "attention.self.value.bias"] = vb
        out[lp + "attention.output.dense.weight"] = p[f"layers.{i}.wo"]
        out[lp + "attention.output.dense.bias"] = p[f"layers.{i}.bo"]
        out[lp + "attention.output.LayerNorm.weight"] = \
            p[f"layers.{i}.ln1_w"]
        out[lp + "attention.output.LayerNorm.bias"] = \
            p[f"layers.{i}.ln1_b"]
        out[lp + "intermediate.dense.weight"] = p[f"layers.{i}.w1"]
        out[lp + "intermediate.dense.bias"] = p[f"layers.{i}.b1"]
        out[lp + "output.dense.weight"] = p[f"layers.{i}.w2"]
        out[lp + "output.dense.bias"] = p[f"layers.{i}.b2"]
        out[lp + "output.LayerNorm.weight"] = p[f"layers.{i}.ln2_w"]
        out[lp + "output.LayerNorm.bias"] = p[f"layers.{i}.ln2_b"]
    return out


def _llama_to_hf(model) -> dict:
    out = {
        "model.embed_tokens.weight": model.tok_emb,
        "model.norm.weight": model.final_norm,
        "lm_head.weight": model.lm_head,
    }
    for i, lyr in enumerate(model.layers):
        lp = f"model.layers.{i}."
        # wq/wk/wv/wgate/wup are row-slice views of the fused params
        out[lp + "self_attn.q_proj.weight"] = lyr.wq
        out[lp + "self_attn.k_proj.weight"] = lyr.wk
        out[lp + "self_attn.v_proj.weight"] = lyr.wv
        out[lp + "self_attn.o_proj.weight"] = lyr.wo
        out[lp + "input_layernorm.weight"] = lyr.in_norm
        out[lp + "post_attention_layernorm.weight"] = lyr.post_norm
        out[lp + "mlp.gate_proj.weight"] = lyr.wgate
        out[lp + "mlp.up_proj.weight"] = lyr.wup
        out[lp + "mlp.down_proj.weight"] = lyr.wdown
    return out


def test_bert_hf_mapping_roundtrip(tmp_path):
    torch.manual_seed(3)
    src = build_model("bert-tiny", device="cpu", dtype=torch.float32)
    _save(tmp_path / "ck", _bert_to_hf(src))
    dst = build_model("bert-tiny", device="cpu", dtype=torch.float32)
    load_weights(dst, tmp_path / "ck")
    ids = torch.randint(4, src.dims.vocab_size, (2, 12))
    mask = torch.ones_like(ids, dtype=torch.int32)
    assert torch.allclose(src(ids, mask), dst(ids, mask), atol=1e-5)


def test_llama_hf_mapping_roundtrip(tmp_path):
    torch.manual_seed(4)
    src = build_model("llama-tiny", device="cpu", dtype=torch.float32)
    _save(tmp_path / "ck", _llama_to_hf(src))
    dst = build_model("llama-tiny", device="cpu", dtype=torch.float32)
    load_weights(dst, tmp_path / "ck")
    ids = torch.randint(4, src.dims.vocab_size, (2, 9))
    mask = torch.ones_like(ids, dtype=torch.int32)
    h1, h2 = src(ids, mask), dst(ids, mask)
    assert torch.allclose(h1, h2, atol=1e-5)
    assert torch.allclose(src.logits(h1[:, -1]), dst.logits(h2[:, -1]),
                          atol=1e-5)


def test_native_checkpoint_passthrough(tmp_path):
    torch.manual_seed(5)
    src = build_model("bert-tiny", device="cpu", dtype=torch.float32)
    _save(tmp_path / "ck", dict(src.named_parameters()))
    dst = build_model("bert-tiny", device="cpu", dtype=torch.float32)
    load_weights(dst, tmp_path / "ck")
    for (n1, p1), (n2, p2) in zip(src.named_parameters(),
                                  dst.named_parameters()):
        assert n1 == n2 and torch.equal(p1, p2)


def test_loader_shape_mismatch_message(tmp_path):
    torch.manual_seed(6)
    src = build_model("bert-tiny", device="cpu", dtype=torch.float32)
    hf = _bert_to_hf(src)
    hf["bert.embeddings.word_embeddings.weight"] = torch.zeros(7, 8)
    _save(tmp_path / "ck", hf)
    dst = build_model("bert-tiny", device="cpu", dtype=torch.float32)
    with pytest.raises(ValueError, match="wrong preset"):
        load_weights(dst, tmp_path / "ck")


def _esm2_to_hf(model) -> dict:
    p = dict(model.named_parameters())
    H = model.dims.hidden_size
    out = {
        "esm.embeddings.word_embeddings.weight": p["word_emb"],
        "esm.encoder.emb_layer_norm_after.weight": p["final_ln_w"],
        "esm.encoder.emb_layer_norm_after.bias": p["final_ln_b"],
    }
    for i in range(len(model.layers)):
        lp = f"esm.encoder.layer.{i}."
        q, k, v = p[f"layers.{i}.wqkv"].split(H, dim=0)
        qb, kb, vb = p[f"layers.{i}.bqkv"].split(H, dim=0)
        out[lp + "attention.self.query.weight"] = q
        out[lp + "attention.self.key.weight"] = k
        out[lp + "attention.self.value.weight"] = v
        out[lp + "attention.self.query.bias"] = qb
        out[lp + "attention.self.key.bias"] = kb
        out[lp + "attention.self.value.bias"] = vb
        out[lp + "attention.output.dense.weight"] = p[f"layers.{i}.wo"]
        out[lp + "attention.output.dense.bias"] = p[f"layers.{i}.bo"]
        out[lp + "attention.LayerNorm.weight"] = p[f"layers.{i}.ln1_w"]
        out[lp + "attention.LayerNorm.bias"] = p[f"layers.{i}.ln1_b"]
        out[lp + "LayerNorm.weight"] = p[f"layers.{i}.ln2_w"]
        out[lp + "LayerNorm.bias"] = p[f"layers.{i}.ln2_b"]
        out[lp + "intermediate.dense.weight"] = p[f"layers.{i}.w1"]
        out[lp + "intermediate.dense.bias"] = p[f"layers.{i}.b1"]
        out[lp + "output.dense.weight"] = p[f"layers.{i}.w2"]
        out[lp + "output.dense.bias"] = p[f"layers.{i}.b2"]
    return out


def test_esm2_hf_mapping_roundtrip(tmp_path):
    torch.manual_seed(7)
    src = build_model("esm2-8m", device="cpu", dtype=torch.float32)
    _save(tmp_path / "ck", _esm2_to_hf(src))
    dst = build_model("esm2-8m", device="cpu", dtype=torch.float32)
    load_weights(dst, tmp_path / "ck")
    ids = torch.randint(4, src.dims.vocab_size, (2, 10))
    mask = torch.ones_like(ids, dtype=torch.int32)
    assert torch.allclose(src(ids, mask), dst(ids, mask), atol=1e-5)
