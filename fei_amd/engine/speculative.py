"""Prompt-lookup speculative decoding (greedy, batch 1).

Agent turns echo their own context heavily — file contents, tool output,
code being edited — so the cheapest possible draft model is the context
itself: match the trailing n-gram against the sequence so far and propose
the tokens that followed its previous occurrence, then verify the whole
block with ONE batched forward through the prefill path (MFMA GEMMs +
flash prefill attention) instead of k sequential decode steps.

Acceptance is exact-greedy: a proposal is accepted iff it equals the
argmax the model produces at that position given the (fully verified)
prefix, so the emitted stream is token-identical to plain greedy decode.
Every verify block emits at least one token (the model's own argmax at
the first mismatch), so progress is guaranteed; KV entries written for
rejected positions are overwritten when decoding reaches them again.

The reference has nothing comparable (its decode is a remote API call);
this is an MI355X-engine feature: the verify block turns k latency-bound
decode steps into one prefill of k+1 tokens.
"""

from __future__ import annotations

from typing import Dict, List, Tuple


class NgramIndex:
    """Incremental index of every n-gram's most recent start position.

    For each n in ``ns`` two dicts are kept: gram -> latest start, and the
    start it displaced (so a lookup can skip the trailing occurrence of
    the query gram itself — which is always the latest one). All updates
    and lookups are O(1) per token; a backward scan of an 8k-token agent
    context in Python would cost ~ms per proposal, comparable to a whole
    decode step.
    """

    def __init__(self, ns: Tuple[int, ...] = (3, 2)):
        self.ns = tuple(sorted(ns, reverse=True))
        self.idx: Dict[int, Dict[Tuple[int, ...], int]] = {n: {} for n in self.ns}
        self.prev: Dict[int, Dict[Tuple[int, ...], int]] = {n: {} for n in self.ns}
        self.ctx: List[int] = []

    def extend(self, tokens: List[int]) -> None:
        for t in tokens:
            self.push(t)

    def push(self, token: int) -> None:
        self.ctx.append(token)
        L = len(self.ctx)
        for n in self.ns:
            if L >= n:
                g = tuple(self.ctx[L - n:])
                d = self.idx[n]
                if g in d:
                    self.prev[n][g] = d[g]
                d[g] = L - n

    def propose(self, k: int) -> List[int]:
        """Up to k draft tokens continuing the trailing n-gram's previous
        occurrence (longest n first). Empty when the context never repeats."""
        ctx = self.ctx
        L = len(ctx)
        for n in self.ns:
            if L <= n:
                continue
            g = tuple(ctx[L - n:])
            s = self.idx[n].get(g)
            if s == L - n:                      # the trailing gram itself
                s = self.prev[n].get(g)
            if s is None:
                continue
            out = ctx[s + n: s + n + k]
            if out:
                return out
        return []
