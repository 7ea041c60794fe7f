"""Tokenizer invariants (server/tokenizer.py), CPU-only.

The streaming property that matters: feeding tokens one at a time through
IncrementalDetokenizer and concatenating the pieces must reproduce the
full-sequence decode exactly — including multibyte UTF-8 runes split
across token boundaries (the naive per-token decode emits U+FFFD)."""

from hypothesis import given, settings
from hypothesis import strategies as st

from arks_amd.server.tokenizer import ByteTokenizer, IncrementalDetokenizer


TOK = ByteTokenizer(vocab_size=512, eos_token_id=2)


@settings(max_examples=300, deadline=None)
@given(text=st.text(max_size=60))
def test_incremental_detok_matches_full_decode(text):
    ids = TOK.encode(text)
    detok = IncrementalDetokenizer(TOK)
    streamed = "".join(detok.feed(t) for t in ids) + detok.flush()
    assert streamed == TOK.decode(ids)


def test_incremental_detok_multibyte_boundary():
    """A 4-byte emoji split across 4 tokens emits nothing until complete,
    then the whole rune at once."""
    ids = TOK.encode("\N{ROCKET}")  # 4 UTF-8 bytes -> 4 tokens
    assert len(ids) == 4
    detok = IncrementalDetokenizer(TOK)
    pieces = [detok.feed(t) for t in ids]
    assert pieces[:3] == ["", "", ""]
    assert pieces[3] == "\N{ROCKET}"
    assert detok.flush() == ""


def test_byte_tokenizer_roundtrip_ascii_and_eos():
    ids = TOK.encode("hello")
    assert TOK.decode(ids + [TOK.eos_token_id]) == "hello"  # eos stripped
    assert TOK.decode([]) == ""


@settings(max_examples=300, deadline=None)
@given(
    text=st.text(alphabet="abcxyz", max_size=40),
    stop=st.text(alphabet="abc", min_size=1, max_size=3),
    cuts=st.lists(st.integers(1, 5), max_size=12),
)
def test_stop_tracker_streaming_equals_batch(text, stop, cuts):
    """The streaming stop tracker (server/api.py StopStringTracker) must emit
    exactly text.split(stop)[0] regardless of how the stream is chunked,
    and never leak any part of the stop string."""
    from arks_amd.server.api import StopStringTracker

    tr = StopStringTracker([stop])
    out, stopped = "", False
    i = 0
    for c in cuts + [len(text)]:
        piece, hit = tr.feed(text[i:i + c])
        out += piece
        i += c
        if hit:
            stopped = True
            break
        if i >= len(text):
            break
    if not stopped:
        out += tr.flush()
    expect = text.split(stop)[0] if stop in text else text
    assert out == expect
    assert stopped == (stop in text)
