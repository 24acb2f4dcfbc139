"""Sentence splitting + sliding-window buffers for semantic chunking.

Replaces the reference's nltk Punkt dependency (nltk is not available
offline) with a deterministic rule-based splitter that keeps the same
*span semantics* the reference relied on (reference
embed/datasets/jsonl_chunk.py:24-58): sentence i covers
[start_i, start_{i+1}) so that ''.join(sentences) == text, and buffers
are sliding windows of +-buffer_size sentences around each sentence.
"""

from __future__ import annotations

import re

# sentence end: ./!/? (+ closing quotes/brackets) followed by whitespace
# and something that looks like a sentence start.
_BOUNDARY = re.compile(
    r"(?<=[.!?])(?:['\")\]]*)\s+(?=[\"'(\[]?[A-Z0-9])"
)

_ABBREV = {
    "dr", "mr", "mrs", "ms", "prof", "sr", "jr", "vs", "etc", "fig",
    "figs", "eq", "eqs", "ref", "refs", "al", "e.g", "i.e", "cf", "ca",
    "no", "vol", "pp", "st", "inc", "ltd", "co", "dept", "univ", "approx",
}


def _is_abbrev_before(text: str, pos: int) -> bool:
    """True if the token ending at pos (exclusive of the period) is a
    known abbreviation or a single initial."""
    j = pos - 1  # index of the '.'
    i = j - 1
    while i >= 0 and (text[i].isalnum() or text[i] == "."):
        i -= 1
    word = text[i + 1 : j].lower()
    if not word:
        return False
    if word in _ABBREV:
        return True
    if len(word) == 1 and word.isalpha():  # single initial "J. Smith"
        return True
    return False


def split_sentences(text: str) -> list[str]:
    """Split text into sentences; ''.join(result) == text."""
    if not text:
        return []
    starts = [0]
    for m in _BOUNDARY.finditer(text):
        period = m.start()  # position right after the [.!?] char
        if text[period - 1] == "." and _is_abbrev_before(text, period):
            continue  # abbreviation, not a sentence end
        starts.append(m.end())  # start of the next sentence
    sentences = []
    for i, s in enumerate(starts):
        e = starts[i + 1] if i + 1 < len(starts) else len(text)
        if e > s:
            sentences.append(text[s:e])
    return sentences


def sentences_to_buffers(sentences: list[str], buffer_size: int) -> list[str]:
    """Sliding window of +-buffer_size sentences around each sentence
    (reference embed/datasets/jsonl_chunk.py:46-58 semantics)."""
    out = []
    n = len(sentences)
    for i in range(n):
        lo = max(0, i - buffer_size)
        hi = min(i + 1 + buffer_size, n)
        out.append("".join(sentences[lo:hi]))
    return out
