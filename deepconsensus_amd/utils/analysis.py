"""Analysis utilities (edit distance, homopolymer content).

Ports the reference's vestigial Beam-era analysis helpers
(model_inference_transforms.py:35-79) as plain utilities, per SURVEY.md
section 7 non-goals.
"""
from __future__ import annotations

from typing import Dict



def edit_distance(a: str, b: str) -> int:
    """Levenshtein distance (iterative DP)."""
    if len(a) < len(b):
        a, b = b, a
    prev = list(range(len(b) + 1))
    for i, ca in enumerate(a, 1):
        cur = [i]
        for j, cb in enumerate(b, 1):
            cur.append(
                min(prev[j] + 1, cur[j - 1] + 1,
                    prev[j - 1] + (ca != cb))
            )
        prev = cur
    return prev[-1]


def homopolymer_content(seq: str, min_run: int = 3) -> float:
    """Fraction of bases inside homopolymer runs of >= min_run."""
    if not seq:
        return 0.0
    in_run = 0
    i = 0
    n = len(seq)
    while i < n:
        j = i
        while j < n and seq[j] == seq[i]:
            j += 1
        if j - i >= min_run:
            in_run += j - i
        i = j
    return in_run / n


def longest_homopolymer(seq: str) -> int:
    best = run = 0
    prev = None
    for ch in seq:
        run = run + 1 if ch == prev else 1
        prev = ch
        best = max(best, run)
    return best


def per_base_error_counts(true_seq: str, pred_seq: str) -> Dict[str, int]:
    """Alignment-free per-position mismatch summary on equal-length strings."""
    n = min(len(true_seq), len(pred_seq))
    out = {"match": 0, "mismatch": 0, "length_diff":
           abs(len(true_seq) - len(pred_seq))}
    for i in range(n):
        if true_seq[i] == pred_seq[i]:
            out["match"] += 1
        else:
            out["mismatch"] += 1
    return out
