"""Encoder strategies: native MI355X models behind the reference's
encoder surface (reference distllm/embed/encoders/*: auto, esm2, esmc —
which delegated to HF transformers / faesm / esm; here the compute is
the in-repo HIP kernel model family).

Config surface parity: `name:` discriminator, pretrained_model_name_or_path,
half_precision/eval-mode options (bf16 is the default compute dtype on
MI355X; fp32 is available for CPU testing).
"""

from __future__ import annotations

from typing import Any, Literal, Union

import torch
from pydantic import Field, model_validator

from ..models import build_model, get_dims
from ..models.config import PRESETS, ModelDims
from ..registry import registry
from ..utils import BaseConfig
from .tokenizers import load_tokenizer

# map common HF model ids used by the reference examples to our presets
_HF_NAME_MAP = {
    "pritamdeka/S-PubMedBert-MS-MARCO": "pubmedbert-base",
    "microsoft/BiomedNLP-PubMedBERT-base-uncased-abstract": "pubmedbert-base",
    "Salesforce/SFR-Embedding-Mistral": "mistral-7b",
    "facebook/esm2_t6_8M_UR50D": "esm2-8m",
    "facebook/esm2_t33_650M_UR50D": "esm2-650m",
    "esmc_300m": "esmc-300m",
    "esmc_600m": "esmc-600m",
}


def resolve_preset(name_or_path: str) -> str:
    if name_or_path in PRESETS:
        return name_or_path
    if name_or_path in _HF_NAME_MAP:
        return _HF_NAME_MAP[name_or_path]
    low = name_or_path.lower()
    if "roberta" in low:
        return "roberta-base"
    if "pubmedbert" in low or "bert" in low:
        return "pubmedbert-base"
    if "esm2" in low:
        return "esm2-650m" if "650" in low else "esm2-8m"
    if "esmc" in low or "cambrian" in low:
        return "esmc-600m" if "600" in low else "esmc-300m"
    if "mistral" in low:
        return "mistral-7b"
    if "llama" in low:
        return "llama3-70b" if "70" in low else "llama3-8b"
    if "qwen" in low:
        return "qwen2-7b"
    raise ValueError(f"cannot resolve model preset for {name_or_path!r}")


class _NativeEncoder:
    """Wraps a native model + tokenizer behind the Encoder protocol:
    encode(batch) -> last_hidden_state [B, S, H] (model dtype)."""

    def __init__(self, dims: ModelDims, device: str, dtype, tokenizer,
                 weights_path: str | None = None):
        self.dims = dims
        self.model = build_model(dims, device=device, dtype=dtype)
        self.tokenizer = tokenizer
        if weights_path:
            from ..models.loader import load_weights

            load_weights(self.model, weights_path)

    @property
    def device(self):
        return self.model.device

    @property
    def dtype(self):
        return self.model.dtype

    @property
    def embedding_size(self) -> int:
        return self.dims.hidden_size

    @torch.no_grad()
    def encode(self, batch_encoding) -> torch.Tensor:
        ids = batch_encoding["input_ids"].to(self.device, non_blocking=True)
        mask = batch_encoding["attention_mask"].to(self.device,
                                                   non_blocking=True)
        return self.model(ids, mask)


def _default_device() -> str:
    return "cuda" if torch.cuda.is_available() else "cpu"


def _ref_tokenizer_alias(data):
    """Reference YAMLs in the wild spell the tokenizer field
    `tokenizer_name` (reference examples/embed/*.yaml; its pydantic
    accepted unknown keys). Map it onto tokenizer_path."""
    if isinstance(data, dict) and "tokenizer_name" in data:
        data = dict(data)  # never mutate the caller's dict
        if not data.get("tokenizer_path"):
            data["tokenizer_path"] = data.pop("tokenizer_name")
        else:
            data.pop("tokenizer_name")
    return data


class AutoEncoderConfig(BaseConfig):
    """Text encoders (BERT-class, Mistral-class)."""

    _ref_alias = model_validator(mode="before")(_ref_tokenizer_alias)

    name: Literal["auto"] = "auto"
    pretrained_model_name_or_path: str = "pubmedbert-base"
    tokenizer_path: str = ""
    weights_path: str = Field(
        "", description="optional local safetensors checkpoint dir"
    )
    half_precision: bool = True  # bf16 on MI355X
    eval_mode: bool = True
    compile_model: bool = False  # N/A: kernels are hand-fused
    quantization: bool = False   # deliberate non-goal on 288 GB HBM


class Esm2EncoderConfig(BaseConfig):
    _ref_alias = model_validator(mode="before")(_ref_tokenizer_alias)

    name: Literal["esm2"] = "esm2"
    pretrained_model_name_or_path: str = "esm2-650m"
    tokenizer_path: str = ""
    weights_path: str = ""
    half_precision: bool = True
    eval_mode: bool = True       # reference esm2.py:27 (always eval)
    compile_model: bool = False  # reference esm2.py:29 (hand-fused)
    faesm: bool = False  # flash attention is always on here


class EsmCambrianEncoderConfig(BaseConfig):
    _ref_alias = model_validator(mode="before")(_ref_tokenizer_alias)

    name: Literal["esmc"] = "esmc"
    pretrained_model_name_or_path: str = "esmc-300m"
    tokenizer_path: str = ""
    weights_path: str = ""
    half_precision: bool = True
    eval_mode: bool = True
    compile_model: bool = False


def _build(cfg) -> _NativeEncoder:
    preset = resolve_preset(cfg.pretrained_model_name_or_path)
    dims = get_dims(preset)
    device = _default_device()
    if device != "cpu" and not cfg.half_precision:
        # The MFMA kernel path is bf16-I/O by design (fp32 accumulate);
        # fp32 end-to-end is the CPU parity mode. Keep the GPU native
        # rather than crashing bf16-only kernels with fp32 tensors.
        print("[encoder] half_precision=False requested on GPU: the "
              "HIP MFMA path is bf16 (fp32 accumulate); running bf16.")
    dtype = torch.bfloat16 if device != "cpu" else torch.float32
    tok = load_tokenizer(
        cfg.tokenizer_path or cfg.pretrained_model_name_or_path,
        arch=dims.arch,
        vocab_size=dims.vocab_size,
        model_max_length=dims.max_position_embeddings,
    )
    return _NativeEncoder(dims, device, dtype, tok,
                          weights_path=cfg.weights_path or None)


EncoderConfigs = Union[
    AutoEncoderConfig, Esm2EncoderConfig, EsmCambrianEncoderConfig
]

STRATEGIES: dict[str, type] = {
    "auto": AutoEncoderConfig,
    "esm2": Esm2EncoderConfig,
    "esmc": EsmCambrianEncoderConfig,
}


def _factory(kwargs: dict[str, Any]):
    name = kwargs.get("name", "")
    if name not in STRATEGIES:
        raise ValueError(
            f"unknown encoder name {name!r}; options: {sorted(STRATEGIES)}"
        )
    return _build(STRATEGIES[name](**kwargs))


def get_encoder(kwargs: dict[str, Any], register: bool = False):
    """Factory; register=True keeps one warm encoder per process
    (reference embed/encoders/__init__.py:34-84 semantics)."""
    if register:
        return registry.get(_factory, dict(kwargs))
    return _factory(dict(kwargs))
