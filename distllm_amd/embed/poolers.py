"""Pooler strategies: mean (excl. BOS/EOS) and last-token, backed by the
fused HIP pooling kernels (pooling.hip). Reference parity:
distllm/embed/poolers/{mean,last_token}.py.

Deliberate divergence: the reference mean pooler's
`attention_mask[:, seq_lengths-1] = 0` vector-index zeroes the column at
EVERY row's length-1 across all rows of a mixed-length batch; we exclude
each row's own last real token only (the intended semantics; identical
for uniform-length batches). See ops/torch_ref.py:mean_pool.
"""

from __future__ import annotations

from typing import Any, Literal, Union

import torch

from .. import ops
from ..utils import BaseConfig


class MeanPoolerConfig(BaseConfig):
    name: Literal["mean"] = "mean"


class MeanPooler:
    def __init__(self, config: MeanPoolerConfig):
        self.config = config

    def pool(self, hidden: torch.Tensor, attention_mask: torch.Tensor,
             normalize: bool = False) -> torch.Tensor:
        """[B,S,H] x [B,S] -> [B,H] fp32 (fused normalize optional)."""
        return ops.mean_pool(hidden, attention_mask, normalize=normalize)


class LastTokenPoolerConfig(BaseConfig):
    name: Literal["last_token"] = "last_token"


class LastTokenPooler:
    def __init__(self, config: LastTokenPoolerConfig):
        self.config = config

    def pool(self, hidden: torch.Tensor, attention_mask: torch.Tensor,
             normalize: bool = False) -> torch.Tensor:
        # left-padding detection as in reference last_token.py:12-38
        left_padded = bool(
            attention_mask[:, -1].sum().item() == attention_mask.shape[0]
        )
        return ops.last_token_pool(hidden, attention_mask,
                                   left_padded=left_padded,
                                   normalize=normalize)


PoolerConfigs = Union[MeanPoolerConfig, LastTokenPoolerConfig]

STRATEGIES: dict[str, tuple[type, type]] = {
    "mean": (MeanPoolerConfig, MeanPooler),
    "last_token": (LastTokenPoolerConfig, LastTokenPooler),
}


def get_pooler(kwargs: dict[str, Any]):
    name = kwargs.get("name", "")
    if name not in STRATEGIES:
        raise ValueError(
            f"unknown pooler name {name!r}; options: {sorted(STRATEGIES)}"
        )
    cfg_cls, impl_cls = STRATEGIES[name]
    return impl_cls(cfg_cls(**kwargs))
