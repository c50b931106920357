"""Repeated-pattern mining over token sequences via a suffix automaton.

Fills the role of the reference's McCreight suffix tree + find_repeat_pattern
(cyliustack/sofa bin/STree.py:51-273: collect substrings occurring exactly
`repeat_limit` times) with a suffix automaton: O(n) construction, every
substring equivalence class carries its occurrence count (endpos size), so
"patterns repeating exactly k times" is a single pass over states.

Works on any hashable token sequence (kernel-name ids, syscall ids).
"""

from __future__ import annotations

from typing import Hashable, List, Sequence, Tuple


class SuffixAutomaton:
    def __init__(self, tokens: Sequence[Hashable]):
        self.n = len(tokens)
        # state arrays
        self.next: List[dict] = [{}]
        self.link: List[int] = [-1]
        self.length: List[int] = [0]
        self.cnt: List[int] = [0]  # endpos size (filled by _count)
        self.first_end: List[int] = [0]  # earliest end position of the class
        last = 0
        for i, ch in enumerate(tokens):
            cur = len(self.next)
            self.next.append({})
            self.link.append(-1)
            self.length.append(self.length[last] + 1)
            self.cnt.append(1)
            self.first_end.append(i + 1)
            p = last
            while p != -1 and ch not in self.next[p]:
                self.next[p][ch] = cur
                p = self.link[p]
            if p == -1:
                self.link[cur] = 0
            else:
                q = self.next[p][ch]
                if self.length[p] + 1 == self.length[q]:
                    self.link[cur] = q
                else:
                    clone = len(self.next)
                    self.next.append(dict(self.next[q]))
                    self.link.append(self.link[q])
                    self.length.append(self.length[p] + 1)
                    self.cnt.append(0)
                    self.first_end.append(self.first_end[q])
                    while p != -1 and self.next[p].get(ch) == q:
                        self.next[p][ch] = clone
                        p = self.link[p]
                    self.link[q] = clone
                    self.link[cur] = clone
            last = cur
        self._count()

    def _count(self) -> None:
        order = sorted(range(1, len(self.length)), key=lambda s: -self.length[s])
        for s in order:
            if self.link[s] > 0:
                self.cnt[self.link[s]] += self.cnt[s]

    def patterns_with_count(self, k: int, tol: int = 0, min_len: int = 2):
        """Yield (start, length, count) of the longest substring of each state
        whose occurrence count is within [k-tol, k+tol]."""
        out = []
        for s in range(1, len(self.length)):
            c = self.cnt[s]
            if k - tol <= c <= k + tol and self.length[s] >= min_len:
                start = self.first_end[s] - self.length[s]
                out.append((start, self.length[s], c))
        return out


def find_repeat_pattern(
    tokens: Sequence[Hashable], k: int, tol: int = 1, min_len: int = 2, max_len: int = 5000
) -> List[Tuple[int, int, int]]:
    """Candidate (start, length, count) patterns repeating ~k times, longest
    first (the reference returns patterns occurring exactly repeat_limit
    times; tol=1 absorbs warmup/cooldown partial iterations)."""
    if len(tokens) < k * min_len:
        return []
    sa = SuffixAutomaton(tokens)
    cands = sa.patterns_with_count(k, tol=tol, min_len=min_len)
    cands = [c for c in cands if c[1] <= max_len]
    cands.sort(key=lambda c: -c[1])
    return cands


def occurrences(tokens: Sequence[Hashable], pattern: Sequence[Hashable]) -> List[int]:
    """Non-overlapping occurrence start indices (left-to-right greedy)."""
    n, m = len(tokens), len(pattern)
    if m == 0 or n < m:
        return []
    out = []
    i = 0
    # simple scan; token comparisons are int==int
    first = pattern[0]
    while i <= n - m:
        if tokens[i] == first and list(tokens[i : i + m]) == list(pattern):
            out.append(i)
            i += m
        else:
            i += 1
    return out
