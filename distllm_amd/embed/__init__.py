"""Embed subsystem: datasets, encoders, poolers, embedders, writers."""

from .datasets import DatasetConfigs, get_dataset  # noqa: F401
from .embedders import EmbedderConfigs, EmbedderResult, get_embedder  # noqa: F401
from .encoders import EncoderConfigs, get_encoder  # noqa: F401
from .poolers import PoolerConfigs, get_pooler  # noqa: F401
from .writers import WriterConfigs, get_writer, unique_shard_dir  # noqa: F401
