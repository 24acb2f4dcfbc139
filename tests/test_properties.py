"""Property-based tests (hypothesis) for the pure-logic contracts the
pipelines lean on: splitter span semantics, batching, varlen block
maps, timer-log round-trip, config YAML round-trip."""

import string

import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from distllm_amd.embed.splitter import (
    sentences_to_buffers,
    split_sentences,
)
from distllm_amd.ops import varlen_block_map
from distllm_amd.utils import BaseConfig, Timer, batch_data, parse_timer_logs

# text with plenty of sentence punctuation and unicode
_text = st.text(
    alphabet=string.ascii_letters + string.digits + " .!?\"'()[]\n\t,;é",
    max_size=400,
)


@settings(deadline=None, max_examples=200)
@given(_text)
def test_split_sentences_spans_cover_text(text):
    """The core contract the semantic chunker relies on
    (reference jsonl_chunk.py span semantics): sentence spans tile the
    input exactly — ''.join(sentences) == text."""
    sents = split_sentences(text)
    assert "".join(sents) == text
    assert all(s for s in sents)  # no empty spans


@settings(deadline=None, max_examples=100)
@given(_text, st.integers(min_value=0, max_value=4))
def test_buffers_shape_and_identity(text, buffer_size):
    sents = split_sentences(text)
    bufs = sentences_to_buffers(sents, buffer_size)
    assert len(bufs) == len(sents)
    # each buffer contains its center sentence
    for i, b in enumerate(bufs):
        assert sents[i] in b
    if buffer_size == 0:
        assert bufs == sents


@settings(deadline=None, max_examples=100)
@given(st.lists(st.integers(), max_size=64),
       st.integers(min_value=1, max_value=17))
def test_batch_data_partitions(data, bs):
    batches = batch_data(data, bs)
    assert [x for b in batches for x in b] == data
    assert all(len(b) <= bs for b in batches)
    assert all(len(b) == bs for b in batches[:-1])


@settings(deadline=None, max_examples=100)
@given(st.lists(st.integers(min_value=0, max_value=300), max_size=16),
       st.sampled_from([16, 64, 128]))
def test_varlen_block_map_properties(lens, qblk):
    cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0))
                      if lens else [0], dtype=torch.int32)
    bs, bq = varlen_block_map(cu, qblk)
    # one block per ceil(len/qblk), zero-length seqs contribute none
    expect = sum((length + qblk - 1) // qblk for length in lens)
    assert len(bs) == len(bq) == expect
    # per-sequence block indices are contiguous 0..n-1 in order
    for s, length in enumerate(lens):
        mine = [int(q) for seq, q in zip(bs, bq) if int(seq) == s]
        assert mine == list(range((length + qblk - 1) // qblk))


@settings(deadline=None, max_examples=50)
@given(st.lists(st.text(alphabet=string.ascii_letters + string.digits,
                        min_size=1, max_size=12), min_size=1, max_size=4))
def test_timer_log_roundtrip(tmp_path_factory, tags):
    log = tmp_path_factory.mktemp("timer") / "t.log"
    with Timer(*tags, logfile=log):
        pass
    stats = parse_timer_logs(log.read_text())
    assert stats[-1].tags == tags
    assert stats[-1].elapsed_s >= 0.0
    assert stats[-1].end_unix >= stats[-1].start_unix


class _Cfg(BaseConfig):
    name: str = "x"
    k: int = 3
    frac: float = 0.5


@settings(deadline=None, max_examples=100)
@given(st.text(alphabet=string.printable, max_size=40).filter(
           lambda s: s == s.strip()),
       st.integers(min_value=-10**9, max_value=10**9),
       st.floats(allow_nan=False, allow_infinity=False, width=32))
def test_config_yaml_roundtrip(tmp_path_factory, name, k, frac):
    p = tmp_path_factory.mktemp("cfg") / "c.yaml"
    _Cfg(name=name, k=k, frac=frac).write_yaml(p)
    back = _Cfg.from_yaml(p)
    assert back.name == name and back.k == k
    assert abs(back.frac - frac) <= 1e-6 * max(1.0, abs(frac))


_jsonish = st.recursive(
    st.none() | st.booleans() | st.integers() | st.floats(
        allow_nan=False) | st.text(max_size=10),
    lambda c: st.lists(c, max_size=3) | st.dictionaries(
        st.text(max_size=8), c, max_size=4),
    max_leaves=12,
)


@settings(deadline=None, max_examples=150)
@given(st.dictionaries(st.text(max_size=12), _jsonish, max_size=6))
def test_config_translators_total_on_junk(data):
    """The reference-dialect translators are total functions over
    arbitrary JSON-ish dicts: they either map or pass through, never
    crash (validation errors belong to pydantic afterwards)."""
    from distllm_amd.embed.encoders import _ref_tokenizer_alias
    from distllm_amd.mcqa.harness import translate_v2_config

    out = translate_v2_config(dict(data))
    assert isinstance(out, dict)
    out2 = _ref_tokenizer_alias(dict(data))
    assert isinstance(out2, dict)
