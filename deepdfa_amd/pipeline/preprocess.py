"""End-to-end preprocessing driver (reference DDFA/scripts/preprocess.sh +
sastvd/scripts/{prepare,getgraphs,dbize,dbize_graphs,abstract_dataflow_full,
dbize_absdf}.py orchestration):

  stage 1  prepare        — dataset dataframe + statement labels
  stage 2  getgraphs      — C file -> CPG (Joern when installed, else the
                            synthetic CPG generator; per-example failures
                            recorded in failed_joern.txt, reference
                            getgraphs.py:57-59)
  stage 3+4+5  dbize      — tables, abstract-dataflow features, train-split
                            vocabulary, graph artifacts

Run: python -m deepdfa_amd.pipeline.preprocess --out storage/processed
     [--sample] [--n 200] [--workers 6]
"""

from __future__ import annotations

import argparse
import logging
import os
from typing import Dict, Set

from ..data.dclass import ds_partition, synthetic_bigvul_df
from .cpg import parse_joern_json, synthetic_cpg
from .dbize import dbize
from .evaluate import get_dep_add_lines_bigvul
from .joern import joern_available

logger = logging.getLogger(__name__)


def prepare(n: int, sample: bool, csv: str = None):
    """Stage 1: the dataset dataframe. With --csv this is the REAL
    MSR_data_cleaned.csv path (reference prepare.py:7-13 semantics via
    data/bigvul_csv.py: comment strip, diff, vul filters); otherwise the
    synthetic Big-Vul-shaped generator."""
    if csv:
        from ..data.bigvul_csv import bigvul_from_csv

        df = bigvul_from_csv(csv, sample=sample)
        if "n_nodes" not in df.columns:
            df = df.assign(n_nodes=df.before.map(lambda s: max(3, len(s.splitlines()))))
    else:
        df = synthetic_bigvul_df(200 if sample else n)
    df = ds_partition(df, "all", split="fixed")
    return df


def getgraphs(df, out_dir: str, workers: int = 1, joern_dir: str = None) -> Dict[int, object]:
    """Stage 2. Priority: pre-exported Joern JSON (--joern-dir holding
    <id>.c.{nodes,edges}.json, the get_func_graph.sc output), then a live
    Joern install, then the synthetic CPG generator. Failures append to
    failed_joern.txt (reference getgraphs.py:57-59)."""
    cpgs = {}
    failed = []
    use_joern = joern_available()
    for _id in df.id:
        try:
            if joern_dir is not None:
                nj = os.path.join(joern_dir, f"{_id}.c.nodes.json")
                ej = os.path.join(joern_dir, f"{_id}.c.edges.json")
                if not (os.path.exists(nj) and os.path.exists(ej)):
                    raise FileNotFoundError(nj)
                cpgs[int(_id)] = parse_joern_json(nj, ej)
            elif use_joern:  # pragma: no cover - needs JVM
                from .joern import run_joern

                cpgs[int(_id)] = run_joern(
                    os.path.join(out_dir, f"{_id}.c"), out_dir
                )
            else:
                cpgs[int(_id)] = synthetic_cpg(int(_id))
        except Exception as e:  # noqa: BLE001 - per-example isolation
            failed.append((int(_id), repr(e)))
    if failed:
        with open(os.path.join(out_dir, "failed_joern.txt"), "a") as f:
            for _id, err in failed:
                f.write(f"{_id}\t{err}\n")
    return cpgs


def statement_labels(df, cpgs, out_dir: str) -> Dict[int, list]:
    """Stage 1b: line-level labels for vulnerable examples
    (statement_labels.pkl contract, reference evaluate.py:239-255). Real
    CSVs carry per-row diff line sets; the synthetic path stands in with
    fixed lines."""
    added: Dict[int, Set[int]] = {}
    has_diff = "added" in df.columns
    for row in df.itertuples():
        if row.vul:
            if has_diff:
                added[int(row.id)] = set(row.added) | set(row.removed)
            else:
                # synthetic stand-in for git-diff added lines
                added[int(row.id)] = {2, 3}
    return get_dep_add_lines_bigvul(
        {i: cpgs[i] for i in added if i in cpgs}, added,
        cache_path=os.path.join(out_dir, "statement_labels.pkl"),
    )


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="storage/processed/bigvul")
    p.add_argument("--n", type=int, default=200)
    p.add_argument("--sample", action="store_true")
    p.add_argument("--workers", type=int, default=1)
    p.add_argument("--csv", default=None,
                   help="real MSR_data_cleaned.csv (reference raw schema)")
    p.add_argument("--joern-dir", default=None,
                   help="directory of pre-exported <id>.c.{nodes,edges}.json")
    p.add_argument(
        "--feat", default="_ABS_DATAFLOW_datatype_all_limitall_1000_limitsubkeys_1000"
    )
    args = p.parse_args(argv)
    logging.basicConfig(level=logging.INFO)
    out = args.out + ("_sample" if args.sample else "")
    os.makedirs(out, exist_ok=True)
    df = prepare(args.n, args.sample, csv=args.csv)
    logger.info("stage 1: %d examples", len(df))
    cpgs = getgraphs(df, out, args.workers, joern_dir=args.joern_dir)
    logger.info("stage 2: %d CPGs", len(cpgs))
    labels = statement_labels(df, cpgs, out)
    logger.info("stage 1b: %d labelled", len(labels))
    vuln_lines = {k: set(v) for k, v in labels.items()}
    train_ids = set(df[df.partition == "train"].id)
    vocabs = dbize(cpgs, out, train_ids, feat=args.feat, vuln_lines=vuln_lines)
    logger.info("stage 3-5: vocab sizes %s", {k: len(v) for k, v in vocabs.items()})
    df.to_parquet(os.path.join(out, "dataset.parquet"))
    return {"out": out, "n": len(df), "vocabs": {k: len(v) for k, v in vocabs.items()}}


if __name__ == "__main__":
    main()
