"""BLEU for code generation evaluation.

Parity target: reference CodeT5/evaluator/ (bleu.py + CodeBLEU's
ngram_match): corpus/sentence BLEU-4 with +1 smoothing as used for the
summarize/translate/refine tasks. Self-contained implementation of the
standard formula (modified n-gram precision, brevity penalty, geometric
mean with add-one smoothing for short sequences).
"""

from __future__ import annotations

import math
from collections import Counter
from typing import List, Sequence


def _ngrams(tokens: Sequence[str], n: int) -> Counter:
    return Counter(tuple(tokens[i : i + n]) for i in range(len(tokens) - n + 1))


def smoothed_bleu4(reference: Sequence[str], candidate: Sequence[str]) -> float:
    """Sentence-level smoothed BLEU-4 (add-1 on numerator/denominator)."""
    if not candidate:
        return 0.0
    log_prec = 0.0
    for n in range(1, 5):
        ref_n = _ngrams(reference, n)
        cand_n = _ngrams(candidate, n)
        overlap = sum(min(c, ref_n.get(g, 0)) for g, c in cand_n.items())
        total = max(1, sum(cand_n.values()))
        log_prec += math.log((overlap + 1.0) / (total + 1.0))
    bp = 1.0
    if len(candidate) < len(reference):
        bp = math.exp(1.0 - len(reference) / max(1, len(candidate)))
    return bp * math.exp(log_prec / 4.0)


def bleu(references: List[Sequence[str]], candidates: List[Sequence[str]]) -> float:
    """Corpus BLEU-4 (sum of clipped counts over the corpus)."""
    overlaps = [0] * 4
    totals = [0] * 4
    ref_len = cand_len = 0
    for ref, cand in zip(references, candidates):
        ref_len += len(ref)
        cand_len += len(cand)
        for n in range(1, 5):
            ref_n = _ngrams(ref, n)
            cand_n = _ngrams(cand, n)
            overlaps[n - 1] += sum(min(c, ref_n.get(g, 0)) for g, c in cand_n.items())
            totals[n - 1] += sum(cand_n.values())
    if cand_len == 0:
        return 0.0
    log_prec = 0.0
    for n in range(4):
        if overlaps[n] == 0:
            log_prec += math.log(1.0 / (2 * max(1, totals[n])))  # standard smoothing
        else:
            log_prec += math.log(overlaps[n] / max(1, totals[n]))
    bp = 1.0 if cand_len > ref_len else math.exp(1.0 - ref_len / max(1, cand_len))
    return bp * math.exp(log_prec / 4.0)
