"""Prompt-lookup speculative decoding: index unit tests + the hard
invariant that speculative greedy decode is token-identical to plain
greedy decode (the acceptance rule IS exact-argmax agreement)."""

import pytest
import torch

from fei_amd.engine.engine import LocalEngine
from fei_amd.engine.speculative import NgramIndex


def brute_propose(ctx, k, ns=(3, 2)):
    L = len(ctx)
    for n in sorted(ns, reverse=True):
        if L <= n:
            continue
        tail = ctx[-n:]
        for i in range(L - n - 1, -1, -1):
            if ctx[i:i + n] == tail:
                out = ctx[i + n: i + n + k]
                if out:
                    return out
    return []


def test_ngram_propose_basic():
    idx = NgramIndex()
    idx.extend([1, 2, 3, 9, 9, 1, 2, 3])
    # trailing (1,2,3) matched at position 0 -> propose what followed: 9,9,...
    assert idx.propose(2) == [9, 9]


def test_ngram_trailing_occurrence_excluded():
    idx = NgramIndex()
    idx.extend([5, 6, 7])
    assert idx.propose(4) == []          # only occurrence is the tail itself


def test_ngram_no_repeat():
    idx = NgramIndex()
    idx.extend(list(range(50)))
    assert idx.propose(8) == []


def test_ngram_matches_brute_force():
    import random
    rng = random.Random(7)
    ctx = [rng.randrange(5) for _ in range(300)]   # small alphabet: repeats
    idx = NgramIndex()
    for i, t in enumerate(ctx):
        idx.push(t)
        if i > 10:
            assert idx.propose(6) == brute_propose(ctx[:i + 1], 6), i


@pytest.fixture(scope="module")
def engine():
    return LocalEngine.create("llama3-tiny")


def _ids(res):
    return res["token_ids"]


def test_spec_equals_plain_on_repetitive_prompt(engine):
    # agent-like context: repeated phrases -> high proposal acceptance
    prompt = ("def add(a, b):\n    return a + b\n" * 6 +
              "def add(a, b):\n    ret")
    plain = engine.generate(prompt, max_new_tokens=48, speculative=False)
    spec = engine.generate(prompt, max_new_tokens=48, speculative=True)
    assert _ids(spec) == _ids(plain)
    assert spec["spec_blocks"] >= 1
    # every block emits at least one token
    assert spec["spec_tokens_per_block"] >= 1.0


def test_spec_equals_plain_on_nonrepetitive_prompt(engine):
    prompt = "The seven distinct colors emerged quickly: "
    plain = engine.generate(prompt, max_new_tokens=24, speculative=False)
    spec = engine.generate(prompt, max_new_tokens=24, speculative=True)
    assert _ids(spec) == _ids(plain)


def test_spec_env_default(engine, monkeypatch):
    monkeypatch.setenv("FEI_SPEC_DECODE", "1")
    out = engine.generate("abc abc abc abc abc ab", max_new_tokens=12)
    assert "spec_blocks" in out


def test_spec_state_resync(engine):
    """After a speculative generate, the device decode state must allow a
    following plain generate (fresh prefill) to behave normally."""
    engine.generate("xyz xyz xyz xyz", max_new_tokens=8, speculative=True)
    out = engine.generate("hello world", max_new_tokens=8, speculative=False)
    assert len(_ids(out)) >= 1
