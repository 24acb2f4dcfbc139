"""Tokenizer providers for the embed/generate pipelines.

The reference used HF AutoTokenizer downloads (embed/datasets/utils.py,
encoders/*.py). This environment has no network, so:
  - a local path with tokenizer files -> HF tokenizers (offline);
  - ESM models -> a native implementation of the fixed ESM alphabet
    (deterministic, identical ids to facebook/esm2_*);
  - otherwise -> a deterministic hash word tokenizer (synthetic bench
    data only — stated in bench output).

All providers expose the HF call surface the pipelines use:
  tok(texts, padding=True, truncation=True, max_length=..,
      return_tensors='pt') -> {input_ids, attention_mask}
"""

from __future__ import annotations

import hashlib
from pathlib import Path

import torch

ESM_VOCAB = [
    "<cls>", "<pad>", "<eos>", "<unk>",
    "L", "A", "G", "V", "S", "E", "R", "T", "I", "D", "P", "K", "Q", "N",
    "F", "Y", "M", "H", "W", "C", "X", "B", "U", "Z", "O", ".", "-",
    "<null_1>", "<mask>",
]


class BatchEncodingDict(dict):
    def to(self, device):
        return BatchEncodingDict(
            {k: v.to(device, non_blocking=True) for k, v in self.items()}
        )


class EsmTokenizer:
    """Exact ESM2 alphabet tokenizer: <cls> seq <eos>, right-padded."""

    def __init__(self, model_max_length: int = 1026):
        self.vocab = {t: i for i, t in enumerate(ESM_VOCAB)}
        self.pad_token_id = self.vocab["<pad>"]
        self.cls_token_id = self.vocab["<cls>"]
        self.eos_token_id = self.vocab["<eos>"]
        self.unk_token_id = self.vocab["<unk>"]
        self.model_max_length = model_max_length
        self.padding_side = "right"
        self.vocab_size = len(ESM_VOCAB)

    def decode(self, ids, skip_special_tokens=True):
        toks = []
        for i in ids:
            t = ESM_VOCAB[int(i)] if 0 <= int(i) < len(ESM_VOCAB) else "<unk>"
            if skip_special_tokens and t.startswith("<"):
                continue
            toks.append(t)
        return "".join(toks)

    def __call__(self, texts, padding=True, truncation=True, max_length=None,
                 return_tensors="pt", pad_to_multiple_of=None, **kw):
        if isinstance(texts, str):
            texts = [texts]
        max_length = max_length or self.model_max_length
        rows = []
        for t in texts:
            ids = [self.cls_token_id]
            for ch in t.strip().upper():
                ids.append(self.vocab.get(ch, self.unk_token_id))
            ids.append(self.eos_token_id)
            if truncation and len(ids) > max_length:
                ids = ids[: max_length - 1] + [self.eos_token_id]
            rows.append(ids)
        return _pad_rows(rows, self.pad_token_id, padding, pad_to_multiple_of)


class HashWordTokenizer:
    """Deterministic synthetic word tokenizer for random-init text
    models: [CLS] w1 .. wn [SEP]; id = stable hash into the vocab."""

    def __init__(self, vocab_size: int = 30528, model_max_length: int = 512):
        self.vocab_size = vocab_size
        self.pad_token_id = 0
        self.cls_token_id = 1
        self.sep_token_id = 2
        self.unk_token_id = 3
        self.model_max_length = model_max_length
        self.padding_side = "right"

    def _wid(self, w: str) -> int:
        h = hashlib.md5(w.lower().encode()).digest()
        return 4 + int.from_bytes(h[:4], "little") % (self.vocab_size - 4)

    @property
    def eos_token_id(self):
        return self.sep_token_id

    def decode(self, ids, skip_special_tokens=True):
        out = []
        for i in ids:
            i = int(i)
            if skip_special_tokens and i < 4:
                continue
            out.append(f"tok{i}")
        return " ".join(out)

    def __call__(self, texts, padding=True, truncation=True, max_length=None,
                 return_tensors="pt", pad_to_multiple_of=None, **kw):
        if isinstance(texts, str):
            texts = [texts]
        max_length = max_length or self.model_max_length
        rows = []
        for t in texts:
            ids = [self.cls_token_id]
            ids += [self._wid(w) for w in t.split()]
            ids.append(self.sep_token_id)
            if truncation and len(ids) > max_length:
                ids = ids[: max_length - 1] + [self.sep_token_id]
            rows.append(ids)
        return _pad_rows(rows, self.pad_token_id, padding, pad_to_multiple_of)


def _pad_rows(rows, pad_id, padding, pad_to_multiple_of):
    width = max((len(r) for r in rows), default=1)
    if pad_to_multiple_of:
        width = (width + pad_to_multiple_of - 1) // pad_to_multiple_of
        width *= pad_to_multiple_of
    ids = torch.full((len(rows), width), pad_id, dtype=torch.long)
    mask = torch.zeros(len(rows), width, dtype=torch.long)
    for i, r in enumerate(rows):
        ids[i, : len(r)] = torch.tensor(r, dtype=torch.long)
        mask[i, : len(r)] = 1
    return BatchEncodingDict(input_ids=ids, attention_mask=mask)


def load_tokenizer(name_or_path: str, arch: str = "bert",
                   vocab_size: int = 30528, model_max_length: int = 512,
                   allow_hash_fallback: bool | None = None):
    """Local HF tokenizer if files exist; native ESM alphabet for protein
    archs; deterministic hash tokenizer otherwise.

    The hash fallback produces meaningless embeddings on real text — it
    exists for synthetic benchmarks and random-weight tests only, so it
    requires explicit opt-in (``allow_hash_fallback=True`` or env
    ``DISTLLM_ALLOW_HASH_TOKENIZER=1``) and emits a loud warning either
    way. Without opt-in, an unrecognized model string raises."""
    import os
    import warnings

    p = Path(name_or_path) if name_or_path else None
    if p is not None and p.exists():
        from transformers import AutoTokenizer

        tok = AutoTokenizer.from_pretrained(str(p))
        return tok
    if arch in ("esm2", "esmc"):
        return EsmTokenizer(model_max_length=model_max_length)
    if allow_hash_fallback is None:
        allow_hash_fallback = os.environ.get(
            "DISTLLM_ALLOW_HASH_TOKENIZER", ""
        ) not in ("", "0")
    if not allow_hash_fallback:
        raise ValueError(
            f"no local tokenizer found at {name_or_path!r} and the "
            "HashWordTokenizer fallback is not enabled. On real text "
            "corpora the hash tokenizer yields meaningless embeddings; "
            "point tokenizer_path at a local HF tokenizer directory, or "
            "opt in for synthetic/bench use with "
            "DISTLLM_ALLOW_HASH_TOKENIZER=1 (or allow_hash_fallback=True)."
        )
    warnings.warn(
        f"[tokenizer] no local tokenizer at {name_or_path!r}: using the "
        "deterministic HashWordTokenizer (synthetic-bench mode). "
        "Embeddings of real text will be meaningless.",
        UserWarning, stacklevel=2,
    )
    return HashWordTokenizer(vocab_size=vocab_size,
                             model_max_length=model_max_length)
