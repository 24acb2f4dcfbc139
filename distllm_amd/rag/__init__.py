"""RAG core: search (HIP flat/binary index), retriever, generator, tasks."""

from .response_synthesizer import RagGenerator  # noqa: F401
from .search import (  # noqa: F401
    BatchedSearchResults,
    HipFlatIndex,
    HipIndexConfig,
    Retriever,
    RetrieverConfig,
)
