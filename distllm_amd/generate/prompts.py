"""Prompt template strategies for the generate pipeline.

Reference parity: distllm/generate/prompts/* (identity, question_chunk,
question_answer, keyword_selection, amp_question):
preprocess(text, contexts?, scores?) -> prompts;
postprocess(responses) -> cleaned outputs.
Templates are re-worded equivalents; postprocess behaviors (first-'?'
sentence extraction, option-number stripping + lowercasing, JSON
merge-back) match the reference.
"""

from __future__ import annotations

import json
import re
from pathlib import Path
from typing import Any, Literal, Union

from pydantic import Field

from ..embed.splitter import split_sentences
from ..utils import BaseConfig


class IdentityPromptTemplateConfig(BaseConfig):
    name: Literal["identity"] = "identity"


class IdentityPromptTemplate:
    def __init__(self, config: IdentityPromptTemplateConfig):
        self.config = config

    def preprocess(self, text, contexts=None, scores=None) -> list[str]:
        return [text] if isinstance(text, str) else list(text)

    def postprocess(self, responses: list[str]) -> list[str]:
        return responses


class QuestionChunkPromptTemplateConfig(BaseConfig):
    name: Literal["question_chunk"] = "question_chunk"


class QuestionChunkPromptTemplate:
    """Generate one deep-understanding question from a text chunk;
    postprocess keeps the first sentence ending in '?'."""

    template = (
        "You are a scientific researcher. Read the text chunk below and "
        "write one high-quality question that requires deep understanding "
        "of the concepts it presents. Avoid questions about paper-specific "
        "details such as results, findings, or references.\n\n"
        "Text: {chunk}\nQuestion:"
    )

    def __init__(self, config: QuestionChunkPromptTemplateConfig):
        self.config = config

    def preprocess(self, text, contexts=None, scores=None) -> list[str]:
        if isinstance(text, str):
            text = [text]
        return [self.template.format(chunk=c) for c in text]

    def postprocess(self, responses: list[str]) -> list[str]:
        out = []
        for r in responses:
            qs = [s for s in split_sentences(r) if s.strip().endswith("?")]
            out.append(qs[0] if qs else "")
        return out


class QuestionAnswerPromptTemplateConfig(BaseConfig):
    name: Literal["question_answer"] = "question_answer"


class QuestionAnswerPromptTemplate:
    """Multiple-choice answering with optional retrieved contexts+scores;
    postprocess strips option numbering and lowercases."""

    template_with_context = (
        "Contexts below are ranked by retrieval score.\n{context}\n"
        "Question: {question}"
        "[INST] Answer the question by choosing exactly one of the "
        "options. Output the chosen option text as-is, with no option "
        "number and no explanation. [/INST]"
        "Answer: "
    )
    template_no_context = (
        "Question: {question}"
        "[INST] Answer the question by choosing exactly one of the "
        "options. Output the chosen option text as-is, with no option "
        "number and no explanation. [/INST]"
        "Answer: "
    )

    def __init__(self, config: QuestionAnswerPromptTemplateConfig):
        self.config = config

    def preprocess(self, text, contexts=None, scores=None) -> list[str]:
        if isinstance(text, str):
            text = [text]
        if contexts is None or scores is None:
            return [self.template_no_context.format(question=q)
                    for q in text]
        prompts = []
        for q, ctx, sc in zip(text, contexts, scores):
            cstr = "\n".join(
                f"Context: {c}, score: {s}" for c, s in zip(ctx, sc)
            )
            prompts.append(self.template_with_context.format(
                context=cstr, question=q))
        return prompts

    _OPT_RE = re.compile(r"^\s*(?:option\s*)?[0-9]+[).:\-]?\s*",
                         re.IGNORECASE)

    def postprocess(self, responses: list[str]) -> list[str]:
        out = []
        for r in responses:
            r = r.strip()
            r = self._OPT_RE.sub("", r)
            out.append(r.lower())
        return out


class KeywordSelectionPromptTemplateConfig(BaseConfig):
    name: Literal["keyword_selection"] = "keyword_selection"
    keywords: Path | list[str] = Field(
        description="keyword list or path to a newline-separated file"
    )


class KeywordSelectionPromptTemplate:
    template = (
        "You are an expert scientist in radiation-based medicine and "
        "biology, well-read in all adjacent domains.\n"
        "Given the keyword list of the domain and a paragraph, select the "
        "3 keywords most relevant to the paragraph, ordered by ascending "
        "relevance.\n"
        "The document:\n\n{document}\n\n----\n\n"
        "List of keywords: {keywords_list}\n\n"
        "Write an answer based on the context.\n"
        "If every keyword is equally irrelevant, return the string "
        "`None of the above` 3 times.\n"
        "Answer: "
    )

    def __init__(self, config: KeywordSelectionPromptTemplateConfig):
        self.config = config
        if isinstance(config.keywords, (str, Path)):
            self.keywords_list = (
                Path(config.keywords).read_text().splitlines()
            )
        else:
            self.keywords_list = list(config.keywords)

    def preprocess(self, text, contexts=None, scores=None) -> list[str]:
        if isinstance(text, str):
            text = [text]
        return [
            self.template.format(document=t,
                                 keywords_list=self.keywords_list)
            for t in text
        ]

    def postprocess(self, responses: list[str]) -> list[str]:
        return [r.strip() for r in responses]


class AMPQuestionPromptTemplateConfig(BaseConfig):
    name: Literal["amp_question"] = "amp_question"


class AMPQuestionPromptTemplate:
    """Protein MCQ generation: input text is a JSON entry; the response
    is merged back into the entry as JSON (reference
    prompts/amp_question.py behavior)."""

    template = (
        "You are an expert in antimicrobial peptides. Using the protein "
        "information below, write one multiple-choice question (4 "
        "options, exactly one correct) testing understanding of the "
        "peptide's function.\n\n"
        "Protein info: {info}\n\n"
        "Return only the question and its options.\nQuestion:"
    )

    def __init__(self, config: AMPQuestionPromptTemplateConfig):
        self.config = config

    def preprocess(self, text, contexts=None, scores=None) -> list[str]:
        if isinstance(text, str):
            text = [text]
        prompts = []
        for t in text:
            try:
                data = json.loads(t)
                info = json.dumps(data, indent=0)
            except (json.JSONDecodeError, TypeError):
                info = t
            prompts.append(self.template.format(info=info))
        return prompts

    def postprocess(self, responses: list[str]) -> list[str]:
        out = []
        for r in responses:
            out.append(json.dumps({"question": r.strip()}))
        return out


PromptConfigs = Union[
    IdentityPromptTemplateConfig,
    QuestionChunkPromptTemplateConfig,
    QuestionAnswerPromptTemplateConfig,
    KeywordSelectionPromptTemplateConfig,
    AMPQuestionPromptTemplateConfig,
]

STRATEGIES: dict[str, tuple[type, type]] = {
    "identity": (IdentityPromptTemplateConfig, IdentityPromptTemplate),
    "question_chunk": (QuestionChunkPromptTemplateConfig,
                       QuestionChunkPromptTemplate),
    "question_answer": (QuestionAnswerPromptTemplateConfig,
                        QuestionAnswerPromptTemplate),
    "keyword_selection": (KeywordSelectionPromptTemplateConfig,
                          KeywordSelectionPromptTemplate),
    "amp_question": (AMPQuestionPromptTemplateConfig,
                     AMPQuestionPromptTemplate),
}


def get_prompt_template(kwargs: dict[str, Any]):
    name = kwargs.get("name", "")
    if name not in STRATEGIES:
        raise ValueError(
            f"unknown prompt name {name!r}; options: {sorted(STRATEGIES)}"
        )
    cfg_cls, impl_cls = STRATEGIES[name]
    return impl_cls(cfg_cls(**kwargs))
