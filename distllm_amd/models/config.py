"""Model architecture configs for the native MI355X model family.

Covers the reference's encoder zoo plus the
decoder family used for generation (K15):
  - BERT-class bidirectional encoders (S-PubMedBERT, pubmedbert,
    bert-base) and RoBERTa-class (offset position ids)
  - ESM2 protein encoders (rotary, no token types)
  - ESMC protein encoders
  - Mistral-class decoder embedders (SFR-Embedding-Mistral)
  - Llama-class causal LMs for generation
  - Qwen2-class causal LMs (llama family + qkv bias)
Dims are from the public model cards; weights are random-init here
(no network) or loadable from local safetensors.
"""

from __future__ import annotations

from typing import Literal

from pydantic import BaseModel


class ModelDims(BaseModel):
    arch: Literal["bert", "esm2", "esmc", "mistral", "llama", "qwen2"]
    vocab_size: int
    hidden_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int | None = None  # None -> MHA
    intermediate_size: int
    max_position_embeddings: int = 512
    layer_norm_eps: float = 1e-12
    rms_norm_eps: float = 1e-5
    rope_theta: float = 10000.0
    type_vocab_size: int = 2          # BERT token types
    gelu: Literal["tanh", "erf"] = "erf"
    tie_word_embeddings: bool = False
    attention_bias: bool = False      # Qwen2: bias on q/k/v only
    # RoBERTa-family: positions = cumsum(ids != pad) + pad (HF
    # create_position_ids_from_input_ids), vs BERT's arange
    position_style: Literal["absolute", "roberta"] = "absolute"
    pad_token_id: int = 0

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_heads

    @property
    def kv_heads(self) -> int:
        return self.num_kv_heads or self.num_heads


PRESETS: dict[str, ModelDims] = {
    # S-PubMedBERT / PubMedBERT-base: the reference's headline embed
    # encoder (examples/embed/AMP.nougat.pubmedbert.yaml)
    "pubmedbert-base": ModelDims(
        arch="bert", vocab_size=30528, hidden_size=768, num_layers=12,
        num_heads=12, intermediate_size=3072, max_position_embeddings=512,
        layer_norm_eps=1e-12, gelu="erf",
    ),
    "bert-tiny": ModelDims(  # test-size
        arch="bert", vocab_size=512, hidden_size=128, num_layers=2,
        num_heads=2, intermediate_size=256, max_position_embeddings=128,
    ),
    # RoBERTa-base (FacebookAI/roberta-base): BERT-family encoder with
    # offset position ids and a single token type
    "roberta-base": ModelDims(
        arch="bert", vocab_size=50265, hidden_size=768, num_layers=12,
        num_heads=12, intermediate_size=3072,
        max_position_embeddings=514, layer_norm_eps=1e-5, gelu="erf",
        type_vocab_size=1, position_style="roberta", pad_token_id=1,
    ),
    # ESM2 sizes (facebook/esm2_t33_650M_UR50D etc.)
    "esm2-8m": ModelDims(
        arch="esm2", vocab_size=33, hidden_size=320, num_layers=6,
        num_heads=20, intermediate_size=1280, max_position_embeddings=1026,
        layer_norm_eps=1e-5, gelu="erf",
    ),
    "esm2-650m": ModelDims(
        arch="esm2", vocab_size=33, hidden_size=1280, num_layers=33,
        num_heads=20, intermediate_size=5120, max_position_embeddings=1026,
        layer_norm_eps=1e-5, gelu="erf",
    ),
    # ESMC (EvolutionaryScale; 300m: H=960 L=30 heads=15, 600m: 1152/36/18)
    "esmc-300m": ModelDims(
        arch="esmc", vocab_size=64, hidden_size=960, num_layers=30,
        num_heads=15, intermediate_size=2560, max_position_embeddings=2048,
        layer_norm_eps=1e-5,
    ),
    "esmc-600m": ModelDims(
        arch="esmc", vocab_size=64, hidden_size=1152, num_layers=36,
        num_heads=18, intermediate_size=3072, max_position_embeddings=2048,
        layer_norm_eps=1e-5,
    ),
    # SFR-Embedding-Mistral (7B): the reference's large embedder
    "mistral-7b": ModelDims(
        arch="mistral", vocab_size=32000, hidden_size=4096, num_layers=32,
        num_heads=32, num_kv_heads=8, intermediate_size=14336,
        max_position_embeddings=32768, rms_norm_eps=1e-5, rope_theta=10000.0,
    ),
    "mistral-tiny": ModelDims(
        arch="mistral", vocab_size=1024, hidden_size=256, num_layers=2,
        num_heads=4, num_kv_heads=2, intermediate_size=512,
        max_position_embeddings=512,
    ),
    # Llama-3-8B for the generate/RAG path (BASELINE config #5)
    "llama3-8b": ModelDims(
        arch="llama", vocab_size=128256, hidden_size=4096, num_layers=32,
        num_heads=32, num_kv_heads=8, intermediate_size=14336,
        max_position_embeddings=8192, rms_norm_eps=1e-5, rope_theta=500000.0,
    ),
    # 140 GB of bf16 weights: fits — with KV headroom — on ONE 288 GB
    # MI355X, no tensor parallelism required (the reference's vLLM
    # configs needed TP8 for far smaller models on 80 GB cards)
    "llama3-70b": ModelDims(
        arch="llama", vocab_size=128256, hidden_size=8192, num_layers=80,
        num_heads=64, num_kv_heads=8, intermediate_size=28672,
        max_position_embeddings=8192, rms_norm_eps=1e-5, rope_theta=500000.0,
    ),
    "llama-tiny": ModelDims(
        arch="llama", vocab_size=2048, hidden_size=256, num_layers=2,
        num_heads=4, num_kv_heads=2, intermediate_size=512,
        max_position_embeddings=512,
    ),
    # Qwen2-7B-Instruct: llama-family + qkv bias (HF Qwen2ForCausalLM)
    "qwen2-7b": ModelDims(
        arch="qwen2", vocab_size=152064, hidden_size=3584, num_layers=28,
        num_heads=28, num_kv_heads=4, intermediate_size=18944,
        max_position_embeddings=32768, rms_norm_eps=1e-6,
        rope_theta=1_000_000.0, attention_bias=True,
    ),
    "qwen2-tiny": ModelDims(
        arch="qwen2", vocab_size=1024, hidden_size=256, num_layers=2,
        num_heads=4, num_kv_heads=2, intermediate_size=512,
        max_position_embeddings=512, rms_norm_eps=1e-6,
        attention_bias=True,
    ),
}


def get_dims(name: str) -> ModelDims:
    if name not in PRESETS:
        raise KeyError(
            f"unknown model preset {name!r}; available: {sorted(PRESETS)}"
        )
    return PRESETS[name]
