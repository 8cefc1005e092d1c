"""C data-flow-graph extraction + the CodeBLEU syntax / dataflow match
components.

Semantics mirror the reference's tree-sitter DFG extractors and match
functions (CodeT5/evaluator/CodeBLEU/parser/DFG.py:11-100 family,
dataflow_match.py:28-147, syntax_match.py:26-75), rebuilt over the native
C parser (cparser.py):

  * DFG items are (var, idx, relation, [parent_vars], [parent_idxs]) with
    relation 'comesFrom' (reads/declarations) or 'computedFrom'
    (assignments);
  * normalize_dataflow renames variables to var_0, var_1, ... in first-use
    order so the match is alpha-invariant;
  * dataflow match = fraction of reference DFG items found in the
    candidate's (with removal); syntax match = fraction of reference AST
    subtrees (s-expressions) present among the candidate's.
"""

from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

from .cparser import Node, all_subtree_sexps, parse_c, remove_comments

DFGItem = Tuple[str, int, str, List[str], List[int]]


def _ident_leaves(node: Node) -> List:
    return [lf for lf in node.leaves() if lf.type == "identifier"]


def _extract(node: Node, states: Dict[str, List[int]], dfg: List[DFGItem]) -> None:
    t = node.type
    if t == "identifier" and node.token is not None:
        code, idx = node.token.text, node.token.idx
        if code in states:
            dfg.append((code, idx, "comesFrom", [code], list(states[code])))
        else:
            states[code] = [idx]
            dfg.append((code, idx, "comesFrom", [], []))
        return
    if t == "init_declarator" and len(node.children) == 2:
        name, value = node.children
        _extract(value, states, dfg)
        value_ids = _ident_leaves(value)
        for nm in _ident_leaves(name):
            code, idx = nm.token.text, nm.token.idx
            dfg.append((code, idx, "comesFrom",
                        [v.token.text for v in value_ids],
                        [v.token.idx for v in value_ids]))
            states[code] = [idx]
        return
    if t in ("assignment_expression", "augmented_assignment") and len(node.children) == 3:
        left, _op, right = node.children
        _extract(right, states, dfg)
        if t == "augmented_assignment":
            _extract(left, states, dfg)  # += reads the left side too
        right_ids = _ident_leaves(right)
        if t == "augmented_assignment":
            right_ids = right_ids + _ident_leaves(left)
        for nm in _ident_leaves(left):
            code, idx = nm.token.text, nm.token.idx
            dfg.append((code, idx, "computedFrom",
                        [v.token.text for v in right_ids],
                        [v.token.idx for v in right_ids]))
            states[code] = [idx]
        return
    if t == "update_expression":  # x++ / ++x: read + redefine
        ids = _ident_leaves(node)
        for nm in ids:
            code, idx = nm.token.text, nm.token.idx
            if code in states:
                dfg.append((code, idx, "computedFrom", [code], list(states[code])))
            else:
                dfg.append((code, idx, "computedFrom", [], []))
            states[code] = [idx]
        return
    if t in ("if_statement", "switch_statement"):
        # branches see the incoming state; the merged outgoing state is the
        # union (reference DFG merges branch states)
        if node.children:
            _extract(node.children[0], states, dfg)  # condition
        branch_states = []
        for child in node.children[1:]:
            st = dict(states)
            _extract(child, st, dfg)
            branch_states.append(st)
        for st in branch_states:
            for k, v in st.items():
                if k in states:
                    states[k] = sorted(set(states[k]) | set(v))
                else:
                    states[k] = v
        return
    if t in ("while_statement", "do_statement", "for_statement"):
        # two passes like the reference loop handling: the second pass sees
        # definitions made inside the body (back-edge flows)
        for _ in range(2):
            for child in node.children:
                _extract(child, states, dfg)
        # a second pass duplicates items — dedup, preserving order
        seen = set()
        uniq = []
        for item in dfg:
            key = (item[0], item[1], item[2], tuple(item[3]), tuple(item[4]))
            if key not in seen:
                seen.add(key)
                uniq.append(item)
        dfg[:] = uniq
        return
    if t == "parameter_declaration":
        for nm in _ident_leaves(node):
            states[nm.token.text] = [nm.token.idx]
            dfg.append((nm.token.text, nm.token.idx, "comesFrom", [], []))
        return
    for child in node.children:
        _extract(child, states, dfg)


def get_data_flow(code: str) -> List[DFGItem]:
    """Extract the DFG; returns [] for unparseable input (never raises)."""
    try:
        root = parse_c(remove_comments(code))
        dfg: List[DFGItem] = []
        _extract(root, {}, dfg)
        dfg.sort(key=lambda x: x[1])
        # keep only items connected to some edge (reference get_data_flow
        # filters to indexed/rooted items)
        return dfg
    except Exception:
        return []


def normalize_dataflow(dataflow: Sequence[DFGItem]):
    """Reference dataflow_match.py:132-147 semantics."""
    var_dict: Dict[str, str] = {}
    i = 0
    out = []
    for var_name, _idx, relationship, par_names, _par_idxs in dataflow:
        for name in par_names:
            if name not in var_dict:
                var_dict[name] = f"var_{i}"
                i += 1
        if var_name not in var_dict:
            var_dict[var_name] = f"var_{i}"
            i += 1
        out.append((var_dict[var_name], relationship,
                    [var_dict[x] for x in par_names]))
    return out


def corpus_dataflow_match(references: List[List[str]], candidates: List[str]) -> float:
    match_count = 0
    total_count = 0
    for refs, cand in zip(references, candidates):
        cand_dfg = normalize_dataflow(get_data_flow(cand))
        for ref in refs:
            ref_dfg = normalize_dataflow(get_data_flow(ref))
            if not ref_dfg:
                continue
            cand_copy = list(cand_dfg)
            total_count += len(ref_dfg)
            for item in ref_dfg:
                if item in cand_copy:
                    match_count += 1
                    cand_copy.remove(item)
    if total_count == 0:
        return 0.0
    return match_count / total_count


def corpus_syntax_match(references: List[List[str]], candidates: List[str]) -> float:
    match_count = 0
    total_count = 0
    for refs, cand in zip(references, candidates):
        cand_sexps = set(all_subtree_sexps(parse_c(remove_comments(cand))))
        for ref in refs:
            ref_sexps = all_subtree_sexps(parse_c(remove_comments(ref)))
            for s in ref_sexps:
                if s in cand_sexps:
                    match_count += 1
            total_count += len(ref_sexps)
    if total_count == 0:
        return 0.0
    return match_count / total_count
