"""Fuzzy string matching on stdlib difflib (fuzzywuzzy is not in the image).

ratio() matches fuzzywuzzy's 0-100 scale so the reference's thresholds
(dedup 80 / scan 90, bin/sofa_aisi.py:241,259; caption matching in
bin/sofa_ml.py:311-341) carry over unchanged.
"""

from __future__ import annotations

from difflib import SequenceMatcher


def ratio(a: str, b: str) -> int:
    return int(round(100 * SequenceMatcher(None, a, b).ratio()))


def best_match(query: str, candidates) -> tuple:
    """Return (best_candidate, score) by ratio."""
    best, best_score = None, -1
    for c in candidates:
        s = ratio(query, c)
        if s > best_score:
            best, best_score = c, s
    return best, best_score
