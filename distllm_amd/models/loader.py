"""Local checkpoint loading (safetensors) with HF name mapping.

Reference parity: the reference loads models by HF name via
AutoModel.from_pretrained (distllm/embed/encoders/auto.py:60-70) and
vLLM's loader; this framework owns the forward, so checkpoints map
onto the native parameter layout instead (TP shards sliced
Megatron-style for tp > 1).

No network in this environment: weights load only from local dirs. The
mapping covers HF BertModel, EsmModel and Llama/Mistral layouts so a
user with a downloaded checkpoint gets real weights; otherwise models
run random-init (the benchmarks declare synthetic data/weights).
"""

from __future__ import annotations

from pathlib import Path

import torch


def _load_safetensors_dir(path: Path) -> dict[str, torch.Tensor]:
    from safetensors.torch import load_file

    files = sorted(Path(path).glob("*.safetensors"))
    if not files:
        raise FileNotFoundError(f"no .safetensors under {path}")
    state: dict[str, torch.Tensor] = {}
    for f in files:
        state.update(load_file(str(f)))
    return state


def _map_bert(src: dict, model) -> dict:
    """HF BertModel / RobertaModel layouts (identical key structure
    under different prefixes)."""
    out = {}
    p = ""
    for cand in ("bert.", "roberta."):
        if any(k.startswith(cand) for k in src):
            p = cand
            break
    out["word_emb"] = src[f"{p}embeddings.word_embeddings.weight"]
    out["pos_emb"] = src[f"{p}embeddings.position_embeddings.weight"]
    out["type_emb"] = src[f"{p}embeddings.token_type_embeddings.weight"]
    out["emb_ln_w"] = src[f"{p}embeddings.LayerNorm.weight"]
    out["emb_ln_b"] = src[f"{p}embeddings.LayerNorm.bias"]
    for i in range(len(model.layers)):
        lp = f"{p}encoder.layer.{i}."
        q_w = src[lp + "attention.self.query.weight"]
        k_w = src[lp + "attention.self.key.weight"]
        v_w = src[lp + "attention.self.value.weight"]
        out[f"layers.{i}.wqkv"] = torch.cat([q_w, k_w, v_w], 0)
        out[f"layers.{i}.bqkv"] = torch.cat(
            [src[lp + "attention.self.query.bias"],
             src[lp + "attention.self.key.bias"],
             src[lp + "attention.self.value.bias"]], 0)
        out[f"layers.{i}.wo"] = src[lp + "attention.output.dense.weight"]
        out[f"layers.{i}.bo"] = src[lp + "attention.output.dense.bias"]
        out[f"layers.{i}.ln1_w"] = src[lp + "attention.output.LayerNorm.weight"]
        out[f"layers.{i}.ln1_b"] = src[lp + "attention.output.LayerNorm.bias"]
        out[f"layers.{i}.w1"] = src[lp + "intermediate.dense.weight"]
        out[f"layers.{i}.b1"] = src[lp + "intermediate.dense.bias"]
        out[f"layers.{i}.w2"] = src[lp + "output.dense.weight"]
        out[f"layers.{i}.b2"] = src[lp + "output.dense.bias"]
        out[f"layers.{i}.ln2_w"] = src[lp + "output.LayerNorm.weight"]
        out[f"layers.{i}.ln2_b"] = src[lp + "output.LayerNorm.bias"]
    return out


def _map_llama(src: dict, model) -> dict:
    out = {}
    p = "model." if any(k.startswith("model.") for k in src) else ""
    out["tok_emb"] = src[f"{p}embed_tokens.weight"]
    out["final_norm"] = src[f"{p}norm.weight"]
    if getattr(model, "lm_head", None) is not None and "lm_head.weight" in src:
        out["lm_head"] = src["lm_head.weight"]
    # tensor-parallel shard slicing (Megatron-style)
    tp = getattr(model, "tp", 1)
    r = getattr(model, "tp_rank", 0)
    d = model.dims
    hd = d.head_dim

    def rows(w, total):
        sh = total // tp
        return w[r * sh : (r + 1) * sh]

    def cols(w, total):
        sh = total // tp
        return w[:, r * sh : (r + 1) * sh]

    for i in range(len(model.layers)):
        lp = f"{p}layers.{i}."
        # fused column-parallel parameters: each part is TP-sliced
        # independently, then concatenated along the output rows
        out[f"layers.{i}.wqkv"] = torch.cat([
            rows(src[lp + "self_attn.q_proj.weight"], d.num_heads * hd),
            rows(src[lp + "self_attn.k_proj.weight"], d.kv_heads * hd),
            rows(src[lp + "self_attn.v_proj.weight"], d.kv_heads * hd),
        ], 0)
        if lp + "self_attn.q_proj.bias" in src:  # Qwen2-family
            out[f"layers.{i}.bqkv"] = torch.cat([
                rows(src[lp + "self_attn.q_proj.bias"], d.num_heads * hd),
                rows(src[lp + "self_attn.k_proj.bias"], d.kv_heads * hd),
                rows(src[lp + "self_attn.v_proj.bias"], d.kv_heads * hd),
            ], 0)
        out[f"layers.{i}.wo"] = cols(src[lp + "self_attn.o_proj.weight"],
                                     d.num_heads * hd)
        out[f"layers.{i}.in_norm"] = src[lp + "input_layernorm.weight"]
        out[f"layers.{i}.post_norm"] = src[
            lp + "post_attention_layernorm.weight"]
        out[f"layers.{i}.wgup"] = torch.cat([
            rows(src[lp + "mlp.gate_proj.weight"], d.intermediate_size),
            rows(src[lp + "mlp.up_proj.weight"], d.intermediate_size),
        ], 0)
        out[f"layers.{i}.wdown"] = cols(src[lp + "mlp.down_proj.weight"],
                                        d.intermediate_size)
    return out


def _ln(src: dict, prefix: str, which: str) -> torch.Tensor:
    """LayerNorm param under either modern (weight/bias) or the legacy
    gamma/beta spelling transformers re-emits for ESM on save."""
    legacy = {"weight": "gamma", "bias": "beta"}[which]
    if f"{prefix}.{which}" in src:
        return src[f"{prefix}.{which}"]
    return src[f"{prefix}.{legacy}"]


def _map_esm2(src: dict, model) -> dict:
    out = {}
    p = "esm." if any(k.startswith("esm.") for k in src) else ""
    out["word_emb"] = src[f"{p}embeddings.word_embeddings.weight"]
    out["final_ln_w"] = _ln(src, f"{p}encoder.emb_layer_norm_after",
                            "weight")
    out["final_ln_b"] = _ln(src, f"{p}encoder.emb_layer_norm_after", "bias")
    for i in range(len(model.layers)):
        lp = f"{p}encoder.layer.{i}."
        out[f"layers.{i}.wqkv"] = torch.cat(
            [src[lp + "attention.self.query.weight"],
             src[lp + "attention.self.key.weight"],
             src[lp + "attention.self.value.weight"]], 0)
        out[f"layers.{i}.bqkv"] = torch.cat(
            [src[lp + "attention.self.query.bias"],
             src[lp + "attention.self.key.bias"],
             src[lp + "attention.self.value.bias"]], 0)
        out[f"layers.{i}.wo"] = src[lp + "attention.output.dense.weight"]
        out[f"layers.{i}.bo"] = src[lp + "attention.output.dense.bias"]
        out[f"layers.{i}.ln1_w"] = _ln(src, lp + "attention.LayerNorm",
                                       "weight")
        out[f"layers.{i}.ln1_b"] = _ln(src, lp + "attention.LayerNorm",
                                       "bias")
        out[f"layers.{i}.ln2_w"] = _ln(src, lp + "LayerNorm", "weight")
        out[f"layers.{i}.ln2_b"] = _ln(src, lp + "LayerNorm", "bias")
        out[f"layers.{i}.w1"] = src[lp + "intermediate.dense.weight"]
        out[f"layers.{i}.b1"] = src[lp + "intermediate.dense.bias"]
        out[f"layers.{i}.w2"] = src[lp + "output.dense.weight"]
        out[f"layers.{i}.b2"] = src[lp + "output.dense.bias"]
    return out


@torch.no_grad()
def load_weights(model, path: str | Path) -> None:
    src = _load_safetensors_dir(Path(path))
    own = dict(model.named_parameters())
    # direct native checkpoint?
    if set(src).issuperset(own):
        mapped = {k: src[k] for k in own}
    else:
        arch = model.dims.arch
        if arch == "bert":
            mapped = _map_bert(src, model)
        elif arch in ("llama", "mistral", "qwen2"):
            mapped = _map_llama(src, model)
        elif arch == "esm2":
            mapped = _map_esm2(src, model)
        else:
            raise ValueError(
                f"no HF mapping for arch {arch}; provide a native checkpoint"
            )
    for name, p in own.items():
        if name in mapped:
            w = mapped[name]
            if tuple(w.shape) != tuple(p.shape):
                raise ValueError(
                    f"checkpoint weight {name} has shape "
                    f"{tuple(w.shape)}, model expects {tuple(p.shape)} — "
                    f"wrong preset for this checkpoint?"
                )
            p.copy_(w.to(p.dtype))
        else:
            print(f"[loader] warning: no weight for {name}, keeping init")
