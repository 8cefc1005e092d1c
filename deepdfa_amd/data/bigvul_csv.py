"""Real Big-Vul (MSR_data_cleaned.csv) loader.

Parity target: reference sastvd/helpers/datasets.py:139-292 (bigvul):
read the raw CSV (id = "Unnamed: 0"), strip comments from
func_before/func_after, compute added/removed line sets + unified diff
(helpers/git.py semantics via difflib here), then apply the reference's
vulnerable-row filters:

  * vul rows with NO added and NO removed lines are dropped;
  * functions with abnormal endings (not '}' / ';', or ');') are dropped;
  * mod_prop = (added+removed)/diff_lines must be < 0.7;
  * before must be > 5 lines;
  * non-vul rows are always kept.

Returns the minimal-column frame [id, before, after, removed, added, diff,
vul, dataset] the rest of the pipeline consumes, with a parquet cache.
"""

from __future__ import annotations

import os
from typing import Optional

import pandas as pd

from ..evaluator.cparser import remove_comments
from ..utils.git import code2diff


def bigvul_from_csv(
    path: str,
    cache_path: Optional[str] = None,
    sample: bool = False,
) -> pd.DataFrame:
    if cache_path and os.path.exists(cache_path):
        try:
            return pd.read_parquet(cache_path)
        except Exception:
            pass
    df = pd.read_csv(path, dtype={"func_before": str, "func_after": str})
    if "Unnamed: 0" in df.columns:
        df = df.rename(columns={"Unnamed: 0": "id"})
    if "id" not in df.columns:
        df = df.reset_index().rename(columns={"index": "id"})
    df["dataset"] = "bigvul"
    if sample:
        df = df.head(200)
    df["vul"] = df["vul"].astype(int)
    df["func_before"] = df["func_before"].fillna("").map(remove_comments)
    df["func_after"] = df["func_after"].fillna("").map(remove_comments)

    diffs = [code2diff(b, a) for b, a in zip(df.func_before, df.func_after)]
    df["added"] = [d["added"] for d in diffs]
    df["removed"] = [d["removed"] for d in diffs]
    df["diff"] = [d["diff"] for d in diffs]
    df["before"] = df["func_before"]
    df["after"] = df["func_after"]

    dfv = df[df.vul == 1]
    if len(dfv):
        dfv = dfv[dfv.apply(lambda x: len(x.added) > 0 or len(x.removed) > 0, axis=1)]
        dfv = dfv[dfv.before.str.strip().str[-1:].isin(["}", ";"])]
        dfv = dfv[~dfv.before.str.strip().str.endswith(");")]

        def mod_prop(x):
            n_diff = max(1, len(x["diff"].splitlines()))
            return (len(x.added) + len(x.removed)) / n_diff

        dfv = dfv[dfv.apply(mod_prop, axis=1) < 0.7]
        dfv = dfv[dfv.before.map(lambda s: len(s.splitlines()) > 5)]
    keep_vuln = set(dfv["id"].tolist())
    df = df[(df.vul == 0) | (df["id"].isin(keep_vuln))].copy()

    out = df[["id", "before", "after", "removed", "added", "diff", "vul", "dataset"]]
    out = out.reset_index(drop=True)
    if cache_path:
        try:
            os.makedirs(os.path.dirname(cache_path), exist_ok=True)
            out.to_parquet(cache_path)
        except Exception:
            pass
    return out
