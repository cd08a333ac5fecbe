"""Property-based tests (hypothesis): parsers and codecs that face
arbitrary user input must never crash and must hold their invariants."""

import string

from hypothesis import given, settings, strategies as st

from fei_amd.engine.tokenizer import ByteTokenizer
from fei_amd.engine.speculative import NgramIndex
from fei_amd.memdir.search import parse_search_args


@settings(max_examples=200, deadline=None)
@given(st.text())
def test_byte_tokenizer_roundtrip(text):
    tok = ByteTokenizer()
    ids = tok.encode(text, add_bos=True, add_eos=True)
    assert ids[0] == tok.bos_id and ids[-1] == tok.eos_id
    assert all(0 <= i < tok.vocab_size for i in ids)
    assert tok.decode(ids) == text


@settings(max_examples=200, deadline=None)
@given(st.text(alphabet=string.printable, max_size=200))
def test_memdir_query_parser_never_crashes(q):
    sq = parse_search_args(q)
    assert sq is not None


@settings(max_examples=100, deadline=None)
@given(st.lists(st.integers(min_value=0, max_value=6), max_size=120),
       st.integers(min_value=1, max_value=10))
def test_ngram_index_matches_brute(ctx, k):
    idx = NgramIndex()
    idx.extend(ctx)
    got = idx.propose(k)
    # brute force oracle (same semantics as tests/test_speculative.py)
    L = len(ctx)
    want = []
    for n in (3, 2):
        if L <= n:
            continue
        tail = ctx[-n:]
        for i in range(L - n - 1, -1, -1):
            if ctx[i:i + n] == tail and ctx[i + n: i + n + k]:
                want = ctx[i + n: i + n + k]
                break
        if want:
            break
    assert got == want


@settings(max_examples=100, deadline=None)
@given(st.lists(st.text(max_size=12), max_size=8))
def test_chat_flatten_never_crashes(contents):
    from fei_amd.serve.api import ChatMessage, flatten_chat
    msgs = [ChatMessage(role="user", content=c) for c in contents]
    out = flatten_chat(msgs)
    assert out.endswith("assistant:")
