"""WER / ExpRate scoring by token-level edit distance.

Capability parity with the reference's ``compute-wer.py`` (SURVEY.md §3.1
#10): DP alignment between predicted and reference token sequences ->
WER = (sub+ins+del)/ref_len; ExpRate = % expressions with zero errors
(plus <=1 / <=2 error rates, the family's extra columns).

Caption-file contract: lines ``"<uid><tab><tok> <tok> ..."``.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Sequence, Tuple


def edit_distance(ref: Sequence[str], hyp: Sequence[str]) -> int:
    """Levenshtein distance over tokens (iterative DP, O(|ref|*|hyp|))."""
    n, m = len(ref), len(hyp)
    if n == 0:
        return m
    if m == 0:
        return n
    prev = list(range(m + 1))
    for i in range(1, n + 1):
        cur = [i] + [0] * m
        for j in range(1, m + 1):
            sub = prev[j - 1] + (ref[i - 1] != hyp[j - 1])
            cur[j] = min(sub, prev[j] + 1, cur[j - 1] + 1)
        prev = cur
    return prev[m]


@dataclass
class WerResult:
    wer: float
    exprate: float
    exprate_le1: float
    exprate_le2: float
    n_expressions: int
    total_errors: int
    total_ref_tokens: int

    def as_dict(self) -> Dict[str, float]:
        return {
            "wer": self.wer, "exprate": self.exprate,
            "exprate_le1": self.exprate_le1, "exprate_le2": self.exprate_le2,
            "n_expressions": self.n_expressions,
        }


def score_pairs(pairs: Sequence[Tuple[Sequence[str], Sequence[str]]]) -> WerResult:
    """pairs of (ref_tokens, hyp_tokens) -> aggregate WER/ExpRate."""
    total_err = 0
    total_ref = 0
    exact = le1 = le2 = 0
    for ref, hyp in pairs:
        d = edit_distance(ref, hyp)
        total_err += d
        total_ref += max(len(ref), 1)
        exact += d == 0
        le1 += d <= 1
        le2 += d <= 2
    n = max(len(pairs), 1)
    return WerResult(
        wer=total_err / max(total_ref, 1),
        exprate=exact / n,
        exprate_le1=le1 / n,
        exprate_le2=le2 / n,
        n_expressions=len(pairs),
        total_errors=total_err,
        total_ref_tokens=total_ref,
    )


def read_caption_file(path: str) -> Dict[str, List[str]]:
    out: Dict[str, List[str]] = {}
    with open(path, "r", encoding="utf-8") as f:
        for line in f:
            parts = line.split()
            if parts:
                out[parts[0]] = parts[1:]
    return out


def score_files(ref_path: str, hyp_path: str) -> WerResult:
    """Score two caption files (reference-compatible CLI contract)."""
    refs = read_caption_file(ref_path)
    hyps = read_caption_file(hyp_path)
    pairs = [(refs[uid], hyps.get(uid, [])) for uid in refs]
    return score_pairs(pairs)


if __name__ == "__main__":  # pragma: no cover
    import sys
    if len(sys.argv) != 3:
        print("usage: python -m wap_amd.eval.wer REF_CAPTIONS HYP_CAPTIONS")
        sys.exit(2)
    r = score_files(sys.argv[1], sys.argv[2])
    print(f"WER {r.wer:.4f}  ExpRate {r.exprate:.4f}  "
          f"<=1 {r.exprate_le1:.4f}  <=2 {r.exprate_le2:.4f}  (n={r.n_expressions})")
