"""RAG generator: retrieve top-k contexts, build the prompt, generate.

Reference parity: distllm/rag/response_synthesizer.py:18-92 (RagGenerator;
retriever=None gives the no-RAG baseline).
"""

from __future__ import annotations

from .search import Retriever


class RagGenerator:
    def __init__(self, generator, retriever: Retriever | None = None):
        self.generator = generator
        self.retriever = retriever

    def generate(self, texts: str | list[str], prompt_template=None,
                 retrieval_top_k: int = 5,
                 retrieval_score_threshold: float = 0.0):
        if isinstance(texts, str):
            texts = [texts]
        if prompt_template is None:
            from ..generate.prompts import IdentityPromptTemplate, \
                IdentityPromptTemplateConfig

            prompt_template = IdentityPromptTemplate(
                IdentityPromptTemplateConfig()
            )

        contexts, scores = None, None
        if self.retriever is not None:
            results, _ = self.retriever.search(
                texts, top_k=retrieval_top_k,
                score_threshold=retrieval_score_threshold,
            )
            contexts = [
                self.retriever.get_texts(idxs)
                for idxs in results.total_indices
            ]
            scores = results.total_scores

        prompts = prompt_template.preprocess(texts, contexts, scores)
        responses = self.generator.generate(prompts)
        return prompt_template.postprocess(responses)
