"""Git-diff helpers: added/removed lines between before/after functions.

Parity target: reference sastvd/helpers/git.py:1-165 (gitdiff via
`git diff --no-index` + unidiff; md_lines; code2diff) — implemented with
difflib (no subprocess/git dependency), same outputs: dicts of added /
removed line numbers and the unified diff text.
"""

from __future__ import annotations

import difflib
from typing import Dict, List


def gitdiff(old: str, new: str) -> str:
    """Unified diff text between two function versions."""
    return "\n".join(
        difflib.unified_diff(old.splitlines(), new.splitlines(), lineterm="", n=0)
    )


def code2diff(old: str, new: str) -> Dict[str, List[int]]:
    """Added (in new) and removed (in old) line numbers (1-based)."""
    added, removed = [], []
    sm = difflib.SequenceMatcher(a=old.splitlines(), b=new.splitlines())
    for tag, i1, i2, j1, j2 in sm.get_opcodes():
        if tag in ("replace", "delete"):
            removed.extend(range(i1 + 1, i2 + 1))
        if tag in ("replace", "insert"):
            added.extend(range(j1 + 1, j2 + 1))
    return {"added": added, "removed": removed, "diff": gitdiff(old, new)}


def allfunc(row) -> Dict:
    """Per-example convenience (reference _c2dhelper semantics): takes a
    mapping with func_before/func_after, returns added/removed/diff."""
    before = row["func_before"] if isinstance(row, dict) else row.func_before
    after = row["func_after"] if isinstance(row, dict) else row.func_after
    return code2diff(before, after)
