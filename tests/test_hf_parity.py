"""Cross-library golden parity vs HuggingFace transformers (CPU fp32).

The reference defines encoder semantics as *whatever HF computes*
(distllm/embed/encoders/auto.py:116-138, esm2.py:107-129); these tests
anchor the native models to that definition independently of our own
torch_ref: build a tiny HF model, export safetensors, load the weights
through models/loader.py, and assert forward equivalence on
mixed-length batches (valid positions only — HF's padded positions are
implementation-defined).
"""

from __future__ import annotations

import pytest
import torch

from distllm_amd.models import build_model
from distllm_amd.models.config import ModelDims
from distllm_amd.models.loader import load_weights

transformers = pytest.importorskip("transformers")

ATOL = 2e-4
RTOL = 2e-4


def _mixed_batch(vocab: int, B=3, S=24, avoid=(), seed=0):
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(4, vocab, (B, S), generator=g)
    for a in avoid:
        ids[ids == a] = 4
    mask = torch.ones(B, S, dtype=torch.long)
    lens = [S, S - 5, S // 2]
    for r, L in enumerate(lens[:B]):
        mask[r, L:] = 0
        ids[r, L:] = 0
    return ids, mask, lens


def _assert_valid_close(ours, theirs, lens, tag):
    for r, L in enumerate(lens):
        a = ours[r, :L].float()
        b = theirs[r, :L].float()
        err = (a - b).abs().max().item()
        scale = b.abs().max().item()
        assert err <= ATOL + RTOL * scale, (
            f"{tag} row {r}: max|diff|={err:.3e} vs scale {scale:.3e}"
        )


def test_bert_matches_hf(tmp_path):
    from transformers import BertConfig, BertModel

    dims = ModelDims(
        arch="bert", vocab_size=256, hidden_size=64, num_layers=3,
        num_heads=4, intermediate_size=128, max_position_embeddings=64,
        layer_norm_eps=1e-12, gelu="erf",
    )
    cfg = BertConfig(
        vocab_size=256, hidden_size=64, num_hidden_layers=3,
        num_attention_heads=4, intermediate_size=128,
        max_position_embeddings=64, type_vocab_size=2,
        layer_norm_eps=1e-12, hidden_act="gelu",
        attn_implementation="eager",
    )
    torch.manual_seed(1)
    hf = BertModel(cfg).eval()
    hf.save_pretrained(tmp_path / "hf", safe_serialization=True)

    ours = build_model(dims, device="cpu", dtype=torch.float32)
    load_weights(ours, tmp_path / "hf")

    ids, mask, lens = _mixed_batch(256, seed=11)
    with torch.no_grad():
        theirs = hf(input_ids=ids, attention_mask=mask).last_hidden_state
        got = ours(ids, mask)
    _assert_valid_close(got, theirs, lens, "bert")


def test_esm2_matches_hf(tmp_path):
    from transformers import EsmConfig, EsmModel

    dims = ModelDims(
        arch="esm2", vocab_size=33, hidden_size=64, num_layers=3,
        num_heads=4, intermediate_size=128, max_position_embeddings=64,
        layer_norm_eps=1e-5,
    )
    cfg = EsmConfig(
        vocab_size=33, hidden_size=64, num_hidden_layers=3,
        num_attention_heads=4, intermediate_size=128,
        max_position_embeddings=64, position_embedding_type="rotary",
        token_dropout=True, emb_layer_norm_before=False,
        layer_norm_eps=1e-5, hidden_act="gelu", pad_token_id=1,
        mask_token_id=32, attn_implementation="eager",
    )
    torch.manual_seed(2)
    hf = EsmModel(cfg, add_pooling_layer=False).eval()
    hf.save_pretrained(tmp_path / "hf", safe_serialization=True)

    ours = build_model(dims, device="cpu", dtype=torch.float32)
    load_weights(ours, tmp_path / "hf")

    # include some <mask> tokens so token-dropout scaling is exercised
    ids, mask, lens = _mixed_batch(32, seed=13)
    ids[0, 3] = 32
    ids[1, 7] = 32
    with torch.no_grad():
        theirs = hf(input_ids=ids, attention_mask=mask).last_hidden_state
        got = ours(ids, mask)
    _assert_valid_close(got, theirs, lens, "esm2")


def test_llama_matches_hf(tmp_path):
    from transformers import LlamaConfig, LlamaModel

    dims = ModelDims(
        arch="llama", vocab_size=128, hidden_size=64, num_layers=3,
        num_heads=4, num_kv_heads=2, intermediate_size=112,
        max_position_embeddings=64, rms_norm_eps=1e-5,
        rope_theta=10000.0,
    )
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, num_hidden_layers=3,
        num_attention_heads=4, num_key_value_heads=2,
        intermediate_size=112, max_position_embeddings=64,
        rms_norm_eps=1e-5, rope_theta=10000.0, attention_bias=False,
        mlp_bias=False, tie_word_embeddings=False,
        attn_implementation="eager",
    )
    torch.manual_seed(3)
    hf = LlamaModel(cfg).eval()
    hf.save_pretrained(tmp_path / "hf", safe_serialization=True)

    ours = build_model(dims, device="cpu", dtype=torch.float32)
    load_weights(ours, tmp_path / "hf")

    # full-length batch (causal; right-padding semantics differ between
    # implementations, last-token pooling uses the valid prefix anyway)
    g = torch.Generator().manual_seed(17)
    ids = torch.randint(4, 128, (2, 24), generator=g)
    mask = torch.ones(2, 24, dtype=torch.long)
    with torch.no_grad():
        theirs = hf(input_ids=ids, attention_mask=mask).last_hidden_state
        got = ours(ids, mask)
    _assert_valid_close(got, theirs, [24, 24], "llama")


def test_mistral_matches_hf(tmp_path):
    from transformers import MistralConfig, MistralModel

    dims = ModelDims(
        arch="mistral", vocab_size=128, hidden_size=64, num_layers=2,
        num_heads=4, num_kv_heads=2, intermediate_size=112,
        max_position_embeddings=64, rms_norm_eps=1e-5,
        rope_theta=10000.0,
    )
    cfg = MistralConfig(
        vocab_size=128, hidden_size=64, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2,
        intermediate_size=112, max_position_embeddings=64,
        rms_norm_eps=1e-5, rope_theta=10000.0,
        tie_word_embeddings=False, attn_implementation="eager",
        sliding_window=None,
    )
    torch.manual_seed(4)
    hf = MistralModel(cfg).eval()
    hf.save_pretrained(tmp_path / "hf", safe_serialization=True)

    ours = build_model(dims, device="cpu", dtype=torch.float32)
    load_weights(ours, tmp_path / "hf")

    g = torch.Generator().manual_seed(19)
    ids = torch.randint(4, 128, (2, 20), generator=g)
    mask = torch.ones(2, 20, dtype=torch.long)
    with torch.no_grad():
        theirs = hf(input_ids=ids, attention_mask=mask).last_hidden_state
        got = ours(ids, mask)
    _assert_valid_close(got, theirs, [20, 20], "mistral")


def test_bert_tokenizer_parity(tmp_path):
    """When a local HF tokenizer dir exists, load_tokenizer must return
    the genuine AutoTokenizer (never the hash fallback)."""
    from transformers import BertTokenizerFast

    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]",
             "hello", "world", "protein", "##s"]
    vf = tmp_path / "vocab.txt"
    vf.write_text("\n".join(vocab))
    tok = BertTokenizerFast(vocab_file=str(vf))
    tok.save_pretrained(tmp_path / "tok")

    from distllm_amd.embed.tokenizers import load_tokenizer

    got = load_tokenizer(str(tmp_path / "tok"), arch="bert")
    enc_a = got(["hello world proteins"], return_tensors="pt")
    enc_b = tok(["hello world proteins"], return_tensors="pt")
    assert torch.equal(enc_a["input_ids"], enc_b["input_ids"])


def test_qwen2_matches_hf(tmp_path):
    """Qwen2 = llama family + bias on the q/k/v projections; the bias
    must survive the fused-wqkv mapping."""
    from transformers import Qwen2Config, Qwen2Model

    dims = ModelDims(
        arch="qwen2", vocab_size=128, hidden_size=64, num_layers=3,
        num_heads=4, num_kv_heads=2, intermediate_size=112,
        max_position_embeddings=64, rms_norm_eps=1e-6,
        rope_theta=10000.0, attention_bias=True,
    )
    cfg = Qwen2Config(
        vocab_size=128, hidden_size=64, num_hidden_layers=3,
        num_attention_heads=4, num_key_value_heads=2,
        intermediate_size=112, max_position_embeddings=64,
        rms_norm_eps=1e-6, rope_theta=10000.0,
        tie_word_embeddings=False, attn_implementation="eager",
        sliding_window=None, use_sliding_window=False,
    )
    torch.manual_seed(5)
    hf = Qwen2Model(cfg).eval()
    # HF inits Linear biases to zero; perturb them so the parity check
    # actually exercises the bias path
    with torch.no_grad():
        for lyr in hf.layers:
            for proj in (lyr.self_attn.q_proj, lyr.self_attn.k_proj,
                         lyr.self_attn.v_proj):
                proj.bias.normal_(std=0.1)
    hf.save_pretrained(tmp_path / "hf", safe_serialization=True)

    ours = build_model(dims, device="cpu", dtype=torch.float32)
    load_weights(ours, tmp_path / "hf")
    # the qkv bias is real data, not zeros
    assert ours.layers[0].bqkv.abs().sum().item() > 0

    g = torch.Generator().manual_seed(23)
    ids = torch.randint(4, 128, (2, 24), generator=g)
    mask = torch.ones(2, 24, dtype=torch.long)
    with torch.no_grad():
        theirs = hf(input_ids=ids, attention_mask=mask).last_hidden_state
        got = ours(ids, mask)
    _assert_valid_close(got, theirs, [24, 24], "qwen2")


def test_roberta_matches_hf(tmp_path):
    """RoBERTa = BERT family + offset position ids derived from the
    pad token (HF create_position_ids_from_input_ids)."""
    from transformers import RobertaConfig, RobertaModel

    dims = ModelDims(
        arch="bert", vocab_size=256, hidden_size=64, num_layers=3,
        num_heads=4, intermediate_size=128, max_position_embeddings=66,
        layer_norm_eps=1e-5, gelu="erf", type_vocab_size=1,
        position_style="roberta", pad_token_id=1,
    )
    cfg = RobertaConfig(
        vocab_size=256, hidden_size=64, num_hidden_layers=3,
        num_attention_heads=4, intermediate_size=128,
        max_position_embeddings=66, type_vocab_size=1,
        layer_norm_eps=1e-5, hidden_act="gelu", pad_token_id=1,
        attn_implementation="eager",
    )
    torch.manual_seed(6)
    hf = RobertaModel(cfg, add_pooling_layer=False).eval()
    hf.save_pretrained(tmp_path / "hf", safe_serialization=True)

    ours = build_model(dims, device="cpu", dtype=torch.float32)
    load_weights(ours, tmp_path / "hf")

    ids, mask, lens = _mixed_batch(256, avoid=(1,), seed=29)
    ids[mask == 0] = 1  # pad token drives RoBERTa position ids
    with torch.no_grad():
        theirs = hf(input_ids=ids, attention_mask=mask).last_hidden_state
        got = ours(ids, mask)
    _assert_valid_close(got, theirs, lens, "roberta")
