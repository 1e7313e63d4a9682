"""Speculative-decoding proposers.

NGramIndex — incremental longest-match-first prompt-lookup (the
suffix-automaton-style upgrade of the naive scan): for each n-gram size it
keeps the LAST TWO occurrence positions of every gram over prompt+output,
updated O(1) per appended token. A proposal is the continuation of the most
recent strictly-earlier occurrence of the longest matching trailing gram —
longer matches first means higher-precision proposals than a fixed n=2
scan, and the incremental index removes the O(context) Python rescan the
round-1 proposer paid per sequence per step.

This matters beyond template-like prompts: greedy decode falls into
attractor cycles (especially visible with random-init weights, but real
models repeat too — lists, code, boilerplate), and those cycles live in
the OUTPUT history, which the index covers at zero extra cost.

Verification stays exact-greedy in the engine: proposals can only change
speed, never tokens.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple


class NGramIndex:
    """Rolling n-gram position index over one sequence's token ids."""

    __slots__ = ("ids", "ns", "maps")

    def __init__(self, ids: Sequence[int] = (), ns: Tuple[int, ...] = (4, 3, 2)):
        self.ids: List[int] = []
        self.ns = tuple(sorted(ns, reverse=True))
        # gram -> (latest start pos, previous start pos or -1)
        self.maps: Dict[int, Dict[tuple, Tuple[int, int]]] = {
            n: {} for n in self.ns
        }
        self.extend(ids)

    def append(self, tok: int) -> None:
        self.ids.append(tok)
        L = len(self.ids)
        for n in self.ns:
            if L >= n:
                gram = tuple(self.ids[L - n:])
                m = self.maps[n]
                prev = m.get(gram)
                m[gram] = (L - n, prev[0] if prev else -1)

    def extend(self, toks: Sequence[int]) -> None:
        for t in toks:
            self.append(t)

    def sync(self, ids: Sequence[int]) -> None:
        """Catch up with the sequence's current ids (append-only growth)."""
        if len(ids) > len(self.ids):
            self.extend(ids[len(self.ids):])

    def propose(self, k: int) -> List[int]:
        """Up to k continuation tokens from the longest matching history."""
        if k <= 0:
            return []
        L = len(self.ids)
        for n in self.ns:
            if L <= n:
                continue
            entry = self.maps[n].get(tuple(self.ids[L - n:]))
            if entry is None:
                continue
            last, prev = entry
            pos = prev if last == L - n else last
            if pos < 0:
                continue
            cont = self.ids[pos + n: pos + n + k]
            if cont:
                return cont
        return []
