"""Preprocessing pipeline + reaching-definitions analysis tests."""

import json
import os

import torch

from deepdfa_amd.analysis import CPG, ReachingDefinitions
from deepdfa_amd.data.dataset import BigVulDatasetLineVD
from deepdfa_amd.data.dclass import synthetic_bigvul_df
from deepdfa_amd.pipeline import (
    dbize,
    get_dataflow_features,
    parse_joern_json,
    synthetic_cpg,
)


def hand_cpg():
    """x = 1; y = x + 2; x = y; cond(x < y)  — known RD solution."""
    nodes = {
        0: {"_label": "METHOD", "name": "f", "lineNumber": 1},
        1: {"_label": "CALL", "name": "<operator>.assignment", "code": "x = 1", "lineNumber": 2},
        2: {"_label": "IDENTIFIER", "name": "x", "order": 1, "typeFullName": "int"},
        3: {"_label": "LITERAL", "name": "1", "code": "1", "order": 2},
        4: {"_label": "CALL", "name": "<operator>.assignment", "code": "y = x + 2", "lineNumber": 3},
        5: {"_label": "IDENTIFIER", "name": "y", "order": 1, "typeFullName": "int"},
        6: {"_label": "CALL", "name": "<operator>.addition", "order": 2},
        7: {"_label": "IDENTIFIER", "name": "x", "order": 1},
        8: {"_label": "LITERAL", "name": "2", "code": "2", "order": 2},
        9: {"_label": "CALL", "name": "<operator>.assignment", "code": "x = y", "lineNumber": 4},
        10: {"_label": "IDENTIFIER", "name": "x", "order": 1, "typeFullName": "int"},
        11: {"_label": "IDENTIFIER", "name": "y", "order": 2},
        12: {"_label": "CALL", "name": "<operator>.lessThan", "code": "x < y", "lineNumber": 5},
        13: {"_label": "IDENTIFIER", "name": "x", "order": 1},
        14: {"_label": "IDENTIFIER", "name": "y", "order": 2},
    }
    edges = [
        (1, 2, "AST"), (1, 3, "AST"), (4, 5, "AST"), (4, 6, "AST"), (6, 7, "AST"),
        (6, 8, "AST"), (9, 10, "AST"), (9, 11, "AST"), (12, 13, "AST"), (12, 14, "AST"),
        (0, 1, "CFG"), (1, 4, "CFG"), (4, 9, "CFG"), (9, 12, "CFG"),
    ]
    return CPG(nodes, edges)


def test_reaching_definitions_known_graph():
    rd = ReachingDefinitions(hand_cpg())
    assert {d.node for d in rd.domain} == {1, 4, 9}
    IN, OUT = rd.solve()
    # after node 1: def(x@1)
    assert {(d.v, d.node) for d in OUT[1]} == {("x", 1)}
    # after node 4: x@1 and y@4
    assert {(d.v, d.node) for d in OUT[4]} == {("x", 1), ("y", 4)}
    # node 9 kills x@1, gens x@9
    assert {(d.v, d.node) for d in OUT[9]} == {("x", 9), ("y", 4)}
    assert {(d.v, d.node) for d in IN[12]} == {("x", 9), ("y", 4)}


def test_rd_loop_fixpoint():
    """Back edge: definitions flow around the loop to the loop head."""
    cpg = hand_cpg()
    cpg.edges.append((12, 1, "CFG"))  # loop back
    IN, OUT = ReachingDefinitions(cpg).solve()
    assert {(d.v, d.node) for d in IN[1]} == {("x", 9), ("y", 4)}


def test_dataflow_features_extraction():
    cpg = hand_cpg()
    df = get_dataflow_features(cpg)
    assert set(df.node_id) == {1, 4, 9}
    row4 = df[df.node_id == 4].iloc[0]
    assert row4.operator == "<operator>.addition"
    assert row4.literal == "2"
    assert row4.datatype == "int"


def test_synthetic_cpg_structure():
    cpg = synthetic_cpg(7)
    rd = ReachingDefinitions(cpg)
    assert len(rd.domain) > 0
    IN, OUT = rd.solve()
    assert all(isinstance(s, set) for s in OUT.values())
    # deterministic
    cpg2 = synthetic_cpg(7)
    assert len(cpg.nodes) == len(cpg2.nodes)


def test_parse_joern_json(tmp_path):
    nodes = [
        {"id": 1, "_label": "METHOD", "name": "f"},
        {"id": 2, "_label": "CALL", "name": "<operator>.assignment", "code": "a = 1"},
        {"id": 3, "_label": "COMMENT", "code": "// nope"},
    ]
    edges = [[1, 2, "CFG"], [1, 2, "CONTAINS"], [1, 3, "AST"]]
    np_, ep_ = str(tmp_path / "n.json"), str(tmp_path / "e.json")
    json.dump(nodes, open(np_, "w"))
    json.dump(edges, open(ep_, "w"))
    cpg = parse_joern_json(np_, ep_)
    assert 3 not in cpg.nodes  # COMMENT dropped
    assert all(t != "CONTAINS" for _, _, t in cpg.edges)


def test_dbize_end_to_end(tmp_path):
    out = str(tmp_path / "processed")
    cpgs = {i: synthetic_cpg(i) for i in range(12)}
    vuln_lines = {0: {3, 4}, 5: {2}}
    vocabs = dbize(cpgs, out, train_ids=range(8), vuln_lines=vuln_lines)
    assert os.path.exists(os.path.join(out, "nodes.csv"))
    assert os.path.exists(os.path.join(out, "edges.csv"))
    assert len(vocabs["operator"]) > 0
    # graph files load into the dataset layer
    from deepdfa_amd.graph import BatchedCFG

    g = BatchedCFG.load(os.path.join(out, "graphs", "0.pt"))
    assert "_ABS_DATAFLOW_datatype" in g.ndata
    assert int(g.ndata["_VULN"].sum()) > 0  # vuln lines marked
    # dataset consumes the artifact dir
    df = synthetic_bigvul_df(12)
    ds = BigVulDatasetLineVD(partition="all", df=df, graph_dir=os.path.join(out, "graphs"))
    gg, extra = ds.item(0)
    assert gg.num_nodes == g.num_nodes
    # self-loops present
    assert gg.num_edges >= gg.num_nodes


def test_preprocess_driver(tmp_path):
    from deepdfa_amd.pipeline import preprocess

    res = preprocess.main(["--out", str(tmp_path / "proc"), "--n", "30"])
    out = res["out"]
    assert os.path.exists(os.path.join(out, "nodes.csv"))
    assert os.path.exists(os.path.join(out, "statement_labels.pkl"))
    assert os.path.exists(os.path.join(out, "dataset.parquet"))
    assert res["vocabs"]["operator"] > 0
    import pickle

    labels = pickle.load(open(os.path.join(out, "statement_labels.pkl"), "rb"))
    assert all(isinstance(v, list) for v in labels.values())


def test_load_graphs_from_csv(tmp_path):
    from deepdfa_amd.pipeline.dbize import load_graphs_from_csv

    out = str(tmp_path / "csvout")
    cpgs = {i: synthetic_cpg(i) for i in range(6)}
    dbize(cpgs, out, train_ids=range(4))
    graphs = load_graphs_from_csv(out)
    assert set(graphs) == set(range(6))
    g = graphs[0]
    assert "_ABS_DATAFLOW_api" in g.ndata
    # matches the baked .pt artifact
    from deepdfa_amd.graph import BatchedCFG

    ref = BatchedCFG.load(os.path.join(out, "graphs", "0.pt"))
    assert torch.equal(g.indices, ref.indices)
    assert torch.equal(g.ndata["_ABS_DATAFLOW_api"], ref.ndata["_ABS_DATAFLOW_api"])


def test_rdg_and_drop_lone_nodes():
    from deepdfa_amd.pipeline.cpg import drop_lone_nodes, rdg, synthetic_cpg

    cpg = synthetic_cpg(7)
    cfg_only = rdg(cpg, "cfg")
    assert cfg_only.edges and all(t == "CFG" for _, _, t in cfg_only.edges)
    ast_only = rdg(cpg, "ast")
    assert all(t == "AST" for _, _, t in ast_only.edges)
    # dropping lone nodes after CFG reduction keeps only CFG-touched nodes
    reduced = drop_lone_nodes(cfg_only)
    touched = {x for s, d, _ in cfg_only.edges for x in (s, d)}
    assert set(reduced.nodes) == touched
    assert len(reduced.nodes) < len(cpg.nodes)


def test_tokenise_lines():
    from deepdfa_amd.data.tokenization import tokenise_lines

    out = tokenise_lines("int fooBar = 1;\nreturn fooBar;")
    assert out == [["int", "foo", "Bar", "1"], ["return", "foo", "Bar"]]


def test_group_nodes_by_line():
    from deepdfa_amd.analysis.dataflow import CPG
    from deepdfa_amd.pipeline.cpg import group_nodes_by_line

    nodes = {
        1: {"_label": "CALL", "lineNumber": 10, "code": "x = a"},
        2: {"_label": "IDENTIFIER", "lineNumber": 10, "code": "x"},
        3: {"_label": "CALL", "lineNumber": 11, "code": "y = x"},
        4: {"_label": "CALL", "lineNumber": -1, "code": "noline"},
    }
    edges = [(1, 2, "AST"), (1, 3, "CFG"), (2, 3, "CFG"), (1, 4, "CFG")]
    g = group_nodes_by_line(CPG(nodes, edges))
    # lines 10 and 11 remain; intra-line edge dropped; duplicate collapsed
    assert set(g.nodes) == {10, 11}
    assert g.edges == [(10, 11, "CFG")]
    assert g.nodes[10]["code"] == "x = a"  # first node per line wins


class TestJoernGraphHelpers:
    """neighbour_nodes / rdg / local-line assignment / dataflow export
    (reference joern.py:372-482 + get_dataflow_output.sc capability)."""

    def _tables(self):
        import pandas as pd

        # 1 -AST-> 2 -AST-> 3; LOCAL 4 under BLOCK 2; 4 -REF-> 5 -EVAL_TYPE-> 6
        nodes = pd.DataFrame([
            {"id": 1, "_label": "METHOD", "name": "f", "lineNumber": 1},
            {"id": 2, "_label": "BLOCK", "name": "", "lineNumber": 1},
            {"id": 3, "_label": "CALL", "name": "<operator>.assignment", "lineNumber": 3},
            {"id": 4, "_label": "LOCAL", "name": "x", "lineNumber": None},
            {"id": 5, "_label": "IDENTIFIER", "name": "x", "lineNumber": 2},
            {"id": 6, "_label": "TYPE", "name": "int", "lineNumber": None},
        ])
        edges = pd.DataFrame([
            {"innode": 1, "outnode": 2, "etype": "AST"},
            {"innode": 2, "outnode": 3, "etype": "AST"},
            {"innode": 2, "outnode": 4, "etype": "AST"},
            {"innode": 4, "outnode": 5, "etype": "REF"},
            {"innode": 5, "outnode": 6, "etype": "EVAL_TYPE"},
            {"innode": 1, "outnode": 3, "etype": "CFG"},
        ])
        return nodes, edges

    def test_rdg_vocabulary(self):
        from deepdfa_amd.pipeline.joern_graph import rdg

        nodes, edges = self._tables()
        assert set(rdg(edges, "ast").etype) == {"AST"}
        assert set(rdg(edges, "reftype").etype) == {"REF", "EVAL_TYPE"}
        assert len(rdg(edges, "cfg")) == 1
        assert len(rdg(edges, "all")) == 5
        import pytest as _pytest
        with _pytest.raises(ValueError):
            rdg(edges, "nope")

    def test_neighbour_nodes_hops(self):
        from deepdfa_amd.pipeline.joern_graph import neighbour_nodes, rdg

        nodes, edges = self._tables()
        one = neighbour_nodes(nodes, rdg(edges, "ast"), [4], 1, False)
        assert one[4] == [2]
        two = neighbour_nodes(nodes, rdg(edges, "reftype"), [4], 2, False)
        assert 6 in two[4]  # LOCAL -> IDENTIFIER -> TYPE
        inter = neighbour_nodes(nodes, rdg(edges, "ast"), [1], 2, True)
        assert 2 in inter[1] and 3 in inter[1]  # hop1 + hop2 accumulated

    def test_assign_line_num_to_local(self):
        from deepdfa_amd.pipeline.joern_graph import assign_line_num_to_local

        nodes, edges = self._tables()
        code = ["int f() {", "  int x;", "  x = 1;", "}"]
        out = assign_line_num_to_local(nodes, edges, code)
        assert out == {4: 2}  # "int x;" is source line 2

    def test_dataflow_solution_export(self, tmp_path):
        import json

        from deepdfa_amd.analysis.dataflow import ReachingDefinitions
        from deepdfa_amd.pipeline.cpg import synthetic_cpg
        from deepdfa_amd.pipeline.joern_graph import export_dataflow_solution
        from deepdfa_amd.pipeline.joern import parse_dataflow_json

        cpg = synthetic_cpg(7)
        path = str(tmp_path / "x.c.dataflow.json")
        out = export_dataflow_solution(cpg, path)
        rd = ReachingDefinitions(cpg)
        sol_in, sol_out = rd.solve()
        # the exported solution matches the Kildall oracle exactly
        some_nonempty = 0
        for n in cpg.nodes:
            rec = out["nodes"][str(n)]
            assert rec["out"] == sorted([d.v, str(d.node)] for d in sol_out[n])
            some_nonempty += bool(rec["out"])
        assert some_nonempty > 3
        # round-trips through the parser (native shape)
        flat = parse_dataflow_json(path)
        assert flat == json.load(open(path))["nodes"]


def test_rd_straightline_monotonicity():
    """Reference property test (dataflow.py:253-317 style): on a
    straight-line chain of fresh-variable assignments, the reaching-def
    set size at each statement grows monotonically with line number (no
    kills, every def survives)."""
    n_stmts = 8
    nodes = {0: {"_label": "METHOD", "name": "f", "lineNumber": 1}}
    edges = []
    nid = 1
    prev = None
    for i in range(n_stmts):
        call = nid
        nodes[call] = {"_label": "CALL", "name": "<operator>.assignment",
                       "code": f"v{i} = {i}", "lineNumber": i + 2}
        nodes[nid + 1] = {"_label": "IDENTIFIER", "name": f"v{i}", "order": 1,
                          "typeFullName": "int"}
        nodes[nid + 2] = {"_label": "LITERAL", "name": str(i), "code": str(i),
                          "order": 2}
        edges += [(call, nid + 1, "AST"), (call, nid + 2, "AST")]
        if prev is not None:
            edges.append((prev, call, "CFG"))
        prev = call
        nid += 3
    cpg = CPG(nodes=nodes, edges=edges)
    rd = ReachingDefinitions(cpg)
    IN, OUT = rd.solve()
    calls = sorted(
        (v["lineNumber"], k) for k, v in nodes.items() if v["_label"] == "CALL"
    )
    sizes = [len(OUT[k]) for _, k in calls]
    assert sizes == list(range(1, n_stmts + 1))  # strictly growing by one


def test_absdf_vocab_determinism_and_bounds():
    """Vocab build (datasets.py:587-692 semantics): deterministic for the
    same train features, capped at limit_all, indices start at 2
    (0 = not-a-definition, 1 = UNKNOWN), most-frequent first."""
    import pandas as pd

    from deepdfa_amd.pipeline.absdf import SUBKEYS, build_vocab, to_hash

    rows = []
    for i in range(60):
        rows.append({k: f"{k}{i % 7}" for k in SUBKEYS})
    df = pd.DataFrame(rows)
    v1 = build_vocab(df, SUBKEYS, limit_all=5)
    v2 = build_vocab(df.copy(), SUBKEYS, limit_all=5)
    assert v1 == v2
    assert len(v1) == 5 and min(v1.values()) == 2 and max(v1.values()) == 6
    # the most frequent hash gets index 2
    top = to_hash(df.iloc[0], SUBKEYS)  # i%7==0 appears most (ceil(60/7)=9x)
    assert v1[top] == 2
    # unlimited: all 7 distinct hashes present
    v_all = build_vocab(df, SUBKEYS, limit_all=1000)
    assert len(v_all) == 7
