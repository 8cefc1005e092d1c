"""CodeBLEU: weighted combination of ngram / weighted-ngram / AST / dataflow
match (reference CodeT5/evaluator/CodeBLEU/calc_code_bleu.py:1-81).

The AST and dataflow components need tree_sitter language parsers; this
environment has none, so those components are gated — when unavailable the
remaining weights are renormalized (and the report says which components
ran). The weighted-ngram component uses the C keyword list (reference
keywords/ directory semantics) with 4x weight on keywords.
"""

from __future__ import annotations

import math
from typing import Dict, List, Sequence

from .bleu import _ngrams, bleu

C_KEYWORDS = {
    "auto", "break", "case", "char", "const", "continue", "default", "do",
    "double", "else", "enum", "extern", "float", "for", "goto", "if", "int",
    "long", "register", "return", "short", "signed", "sizeof", "static",
    "struct", "switch", "typedef", "union", "unsigned", "void", "volatile",
    "while",
}


def weighted_ngram_match(references: List[Sequence[str]],
                         candidates: List[Sequence[str]], kw_weight=4.0) -> float:
    overl = [0.0] * 4
    total = [0.0] * 4
    for ref, cand in zip(references, candidates):
        for n in range(1, 5):
            ref_n = _ngrams(ref, n)
            cand_n = _ngrams(cand, n)

            def w(g):
                return kw_weight if any(t in C_KEYWORDS for t in g) else 1.0

            overl[n - 1] += sum(min(c, ref_n.get(g, 0)) * w(g) for g, c in cand_n.items())
            total[n - 1] += sum(c * w(g) for g, c in cand_n.items())
    log_p = 0.0
    for n in range(4):
        p = (overl[n] + 1.0) / (total[n] + 1.0)
        log_p += math.log(p)
    return math.exp(log_p / 4.0)


def _tree_sitter_available() -> bool:
    try:
        import tree_sitter  # noqa: F401

        return True
    except ImportError:
        return False


def syntax_match(references, candidates) -> float:  # pragma: no cover - gated
    raise RuntimeError("syntax_match needs tree_sitter (not installed)")


def dataflow_match(references, candidates) -> float:  # pragma: no cover - gated
    raise RuntimeError("dataflow_match needs tree_sitter (not installed)")


def calc_code_bleu(
    references: List[str], candidates: List[str],
    weights=(0.25, 0.25, 0.25, 0.25),
) -> Dict[str, float]:
    refs = [r.split() for r in references]
    cands = [c.split() for c in candidates]
    comps = {
        "ngram_match": bleu(refs, cands),
        "weighted_ngram_match": weighted_ngram_match(refs, cands),
    }
    used_w = [weights[0], weights[1]]
    if _tree_sitter_available():  # pragma: no cover
        comps["syntax_match"] = syntax_match(references, candidates)
        comps["dataflow_match"] = dataflow_match(references, candidates)
        used_w += [weights[2], weights[3]]
    total_w = sum(used_w)
    score = sum(w / total_w * v for w, v in zip(used_w, comps.values()))
    comps["code_bleu"] = score
    comps["components_used"] = len(used_w)
    return comps
