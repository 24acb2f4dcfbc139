"""Native MI355X model family + factory."""

from __future__ import annotations

import torch

from .bert import BertEncoder
from .config import PRESETS, ModelDims, get_dims
from .esm import Esm2Encoder, EsmCEncoder
from .mistral import DecoderModel

__all__ = [
    "BertEncoder",
    "DecoderModel",
    "Esm2Encoder",
    "EsmCEncoder",
    "ModelDims",
    "PRESETS",
    "build_model",
    "get_dims",
]


def build_model(name_or_dims: str | ModelDims, device: str | None = None,
                dtype=torch.bfloat16, **kw):
    dims = (
        get_dims(name_or_dims)
        if isinstance(name_or_dims, str)
        else name_or_dims
    )
    if dims.arch == "bert":
        m = BertEncoder(dims, dtype=dtype, **kw)
    elif dims.arch == "esm2":
        m = Esm2Encoder(dims, dtype=dtype, **kw)
    elif dims.arch == "esmc":
        m = EsmCEncoder(dims, dtype=dtype, **kw)
    elif dims.arch in ("mistral", "llama", "qwen2"):
        m = DecoderModel(dims, dtype=dtype, **kw)
    else:
        raise ValueError(f"unknown arch {dims.arch}")
    if device is not None:
        m = m.to(device)
    m.eval()
    return m
