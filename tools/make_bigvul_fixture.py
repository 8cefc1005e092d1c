"""Generate the frozen Big-Vul fixture (tests/fixtures/bigvul/):

  * MSR_data_cleaned.csv — 200 rows in the REAL raw schema (all columns the
    reference's dtype map names, sastvd/helpers/datasets.py:160-196),
    with deliberate edge cases: block/line comments inside functions,
    embedded commas/quotes/newlines (CSV quoting), a vulnerable row with
    no diff, an abnormal-ending function, and a too-short function (all
    three must be FILTERED by the loader);
  * joern/<id>.c.nodes.json / .edges.json — Joern-export-shaped CPG JSON
    (get_func_graph.sc format) for every kept id.

Run once; the output is committed so tests are hermetic:
    python tools/make_bigvul_fixture.py
"""

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd

from deepdfa_amd.data.text_dataset import synthetic_func_source
from deepdfa_amd.pipeline.cpg import synthetic_cpg

OUT = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "tests", "fixtures", "bigvul")


def patched_version(func: str) -> str:
    """Remove the strcpy line and add a bounds check (a plausible fix)."""
    lines = [l for l in func.split("\n") if "strcpy(" not in l]
    for i, l in enumerate(lines):
        if l.strip().startswith("memcpy("):
            lines.insert(i, "  if (!check_bounds(buf, len)) return -1;")
            break
    return "\n".join(lines)


def main():
    os.makedirs(os.path.join(OUT, "joern"), exist_ok=True)
    rng = np.random.RandomState(42)
    rows = []
    n = 200
    for i in range(n):
        vul = int(rng.rand() < 0.25)
        func = synthetic_func_source(i, n_lines=8 + rng.randint(10), vul=vul)
        if i % 17 == 0:  # comments the loader must strip
            func = "/* refactor\n  helper, v2 */\n" + func.replace(
                "return 0;", "return 0; // ok path")
        after = patched_version(func) if vul else func
        summary = f'Heap overflow, "quoted", with, commas #{i}'
        if vul and i % 31 == 0:
            after = func  # no-diff vul row -> must be filtered
        rows.append({
            "Unnamed: 0": i,
            "Access Gained": "None", "Attack Origin": "Remote",
            "Authentication Required": "Not required", "Availability": "Partial",
            "CVE ID": f"CVE-2018-{10000+i}", "CVE Page": "https://example/cve",
            "CWE ID": "CWE-119", "Complexity": "Low",
            "Confidentiality": "Partial", "Integrity": "Partial",
            "Known Exploits": "", "Publish Date": "2018-01-02",
            "Update Date": "2019-03-04", "Score": float(rng.rand() * 10),
            "Summary": summary, "Vulnerability Classification": "Overflow",
            "add_lines": 1, "codeLink": "https://example/commit",
            "commit_id": f"{i:040x}", "commit_message": "fix bounds\ncheck",
            "del_lines": 1, "file_name": f"src/mod_{i % 7}.c",
            "files_changed": "1", "func_after": after, "func_before": func,
            "lang": "C", "lines_after": "", "lines_before": "",
            "parentID": f"{i + 1:040x}", "patch": "@@ -1 +1 @@",
            "project": f"proj{i % 9}", "project_after": "", "project_before": "",
            "vul": vul, "vul_func_with_fix": after if vul else "",
        })
    # hand-made filter cases
    rows[3]["vul"] = 1
    rows[3]["func_before"] = "int f(int a)\n{\n  return a\n"  # abnormal ending
    rows[3]["func_after"] = "int g(int a)\n{\n  return a;\n}"
    rows[7]["vul"] = 1
    rows[7]["func_before"] = "short_fn();"  # too short (< 6 lines)
    rows[7]["func_after"] = "short_fn(1);"
    df = pd.DataFrame(rows)
    df.to_csv(os.path.join(OUT, "MSR_data_cleaned.csv"), index=False)

    for i in range(n):
        cpg = synthetic_cpg(i)
        nodes = [[nid, props] for nid, props in cpg.nodes.items()]
        edges = [[s, d, t] for s, d, t in cpg.edges]
        with open(os.path.join(OUT, "joern", f"{i}.c.nodes.json"), "w") as f:
            json.dump(nodes, f)
        with open(os.path.join(OUT, "joern", f"{i}.c.edges.json"), "w") as f:
            json.dump(edges, f)
    print(f"fixture written to {OUT}: {n} rows")


if __name__ == "__main__":
    main()
