#!/usr/bin/env python3
"""Aggregate profiledata.jsonl / timedata.jsonl to GFLOPs and ms/example
(reference scripts/report_profiling.py:18-66 contract)."""

import json
import os
import sys


def main(run_dir="."):
    prof = os.path.join(run_dir, "profiledata.jsonl")
    timef = os.path.join(run_dir, "timedata.jsonl")
    if os.path.exists(prof):
        rows = [json.loads(l) for l in open(prof) if l.strip()]
        total_flops = sum(r["flops"] for r in rows)
        total_macs = sum(r["macs"] for r in rows)
        n = sum(r["batch_size"] for r in rows)
        print(f"gflops total: {total_flops/1e9:.1f}")
        print(f"gmacs total: {total_macs/1e9:.1f}")
        print(f"gflops avg/example: {total_flops/1e9/max(1,n):.4f}")
    if os.path.exists(timef):
        rows = [json.loads(l) for l in open(timef) if l.strip()]
        total_ms = sum(r["time_ms"] for r in rows)
        n = sum(r["batch_size"] for r in rows)
        print(f"ms/example: {total_ms/max(1,n):.3f}")


if __name__ == "__main__":
    main(sys.argv[1] if len(sys.argv) > 1 else ".")
