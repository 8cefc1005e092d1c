"""CodeT5 extras: clone model, generation + beam, CodeBLEU, run_exp matrix."""

import torch

from deepdfa_amd.evaluator import bleu, calc_code_bleu, smoothed_bleu4
from deepdfa_amd.models.codet5 import CloneModel
from deepdfa_amd.models.t5 import T5Config
from deepdfa_amd.train import run_clone, run_gen


def small_cfg():
    return T5Config(vocab_size=300, d_model=64, d_kv=16, d_ff=128,
                    num_layers=1, num_decoder_layers=1, num_heads=4)


def test_clone_model_forward():
    torch.manual_seed(0)
    model = CloneModel(config=small_cfg(), max_source_length=20)
    ids = torch.randint(3, 300, (2, 40))
    ids[:, 18] = 2
    ids[:, 19] = 0
    ids[:, 38] = 2
    ids[:, 39] = 0
    loss, prob = model(ids, labels=torch.tensor([0, 1]))
    assert prob.shape == (2, 2)
    loss.backward()


def test_generate_greedy_and_beam():
    torch.manual_seed(0)
    from deepdfa_amd.models.t5 import T5ForConditionalGeneration

    m = T5ForConditionalGeneration(small_cfg()).eval()
    ids = torch.randint(3, 300, (2, 12))
    out_g = m.generate(ids, max_length=6, num_beams=1)
    assert out_g.shape[0] == 2 and out_g.shape[1] <= 6
    out_b = m.generate(ids, max_length=6, num_beams=3)
    assert out_b.shape[0] == 2


def test_bleu_properties():
    ref = "int main ( ) { return 0 ; }".split()
    assert smoothed_bleu4(ref, ref) > 0.95
    assert smoothed_bleu4(ref, "totally different words here".split()) < 0.2
    assert bleu([ref], [ref]) > 0.95
    r = calc_code_bleu(["int x = 1 ;"], ["int x = 1 ;"])
    assert r["code_bleu"] > 0.9
    assert r["components_used"] == 4  # native AST/DFG components
    # keyword weighting: a keyword MATCH outweighs an identifier match at
    # the same number of token mismatches
    kw_match = calc_code_bleu(["return x ;"], ["return y ;"])
    kw_miss = calc_code_bleu(["return x ;"], ["goto x ;"])
    assert kw_match["weighted_ngram_match"] > kw_miss["weighted_ngram_match"]


def test_run_gen_end_to_end(tmp_path):
    res = run_gen.main([
        "--do_train", "--do_test", "--num_train_epochs", "1",
        "--n_synthetic", "16", "--num_layers", "1", "--d_model", "64",
        "--max_source_length", "48", "--max_target_length", "12",
        "--beam_size", "2", "--output_dir", str(tmp_path / "gen"),
    ])
    assert "train_loss" in res and "bleu4" in res
    assert 0.0 <= res["bleu4"] <= 1.0


def test_run_clone_end_to_end(tmp_path):
    res = run_clone.main([
        "--do_train", "--do_test", "--num_train_epochs", "1",
        "--n_synthetic", "16", "--num_layers", "1", "--d_model", "64",
        "--max_source_length", "32", "--output_dir", str(tmp_path / "clone"),
    ])
    assert "train_loss" in res and "test_acc" in res


class TestCodeBLEUFull:
    """AST + dataflow CodeBLEU components over the native C parser
    (reference CodeT5/evaluator/CodeBLEU parity, VERDICT round-1 item 5)."""

    REF = "int f ( int n ) { int a = n + 1 ; if ( a > 2 ) { a = a * 2 ; } return a ; }"

    def test_identical_scores_one(self):
        from deepdfa_amd.evaluator.calc_code_bleu import calc_code_bleu

        out = calc_code_bleu([self.REF], [self.REF])
        assert out["components_used"] == 4
        for k in ("ngram_match", "weighted_ngram_match", "syntax_match",
                  "dataflow_match", "code_bleu"):
            assert abs(out[k] - 1.0) < 1e-9, (k, out[k])

    def test_alpha_renaming_invariance(self):
        """Renamed variables keep syntax and dataflow at 1.0 while ngram
        drops — the defining CodeBLEU property."""
        from deepdfa_amd.evaluator.calc_code_bleu import calc_code_bleu

        cand = self.REF.replace(" a ", " z ").replace(" n ", " m ")
        out = calc_code_bleu([self.REF], [cand])
        assert out["syntax_match"] == 1.0
        assert out["dataflow_match"] == 1.0
        assert out["ngram_match"] < 1.0
        assert out["code_bleu"] < 1.0

    def test_structural_difference_detected(self):
        from deepdfa_amd.evaluator.calc_code_bleu import calc_code_bleu

        out = calc_code_bleu([self.REF], ["void g ( ) { while ( x ) y ++ ; }"])
        assert out["syntax_match"] < 0.3
        assert out["code_bleu"] < 0.3

    def test_parser_never_raises_on_garbage(self):
        from deepdfa_amd.evaluator.cparser import parse_c
        from deepdfa_amd.evaluator.dfg_c import get_data_flow

        for junk in ["", "@@@ %% ((", "int int int", "if ( { ) }", "a +", "}{"]:
            root = parse_c(junk)
            assert root.type == "translation_unit"
            get_data_flow(junk)  # must not raise

    def test_dfg_extraction_semantics(self):
        from deepdfa_amd.evaluator.dfg_c import get_data_flow

        dfg = get_data_flow("int a = b ; a = a + c ;")
        # b read (new), a comesFrom b, a/c reads, a computedFrom a,c
        rels = [(v, r, tuple(ps)) for v, _i, r, ps, _pi in dfg]
        assert ("a", "comesFrom", ("b",)) in rels
        assert any(v == "a" and r == "computedFrom" and "c" in ps
                   for v, r, ps in rels)

    def test_dfg_loop_backedge(self):
        from deepdfa_amd.evaluator.dfg_c import get_data_flow

        dfg = get_data_flow("while ( i < n ) { i = i + 1 ; }")
        # the second pass sees i defined inside the body (back-edge flow)
        computed = [x for x in dfg if x[2] == "computedFrom" and x[0] == "i"]
        assert computed

    def test_syntax_match_partial(self):
        from deepdfa_amd.evaluator.dfg_c import corpus_syntax_match

        score = corpus_syntax_match(
            [["int f ( ) { return a + b ; }"]],
            ["int g ( ) { int x ; return a + b ; }"],
        )
        assert 0.0 < score < 1.0


def test_cparser_never_raises_fuzz():
    """The tolerant parser must survive arbitrary input (reference behavior:
    tree-sitter always yields a tree; calc_code_bleu feeds it raw model
    output). Random printable garbage, truncated C, unbalanced brackets."""
    import random
    import string

    from deepdfa_amd.evaluator.cparser import all_subtree_sexps, parse_c

    rng = random.Random(0)
    samples = [
        "", "{", "}}}}", "int f( {", "a+++++b;", '"unterminated',
        "/* unclosed", "int x = ;;;", "\x00\x01\x02",
    ]
    for _ in range(40):
        n = rng.randint(1, 120)
        samples.append("".join(rng.choice(string.printable) for _ in range(n)))
    chunks = ["int f(int a){", "return a+1;", "}", "for(;;)", "while(1)",
              "x[3] =", "struct s {", "((", "))", "case 1:", "->y", "*p",
              '"str"', "'c'", "0x1f", "e?a:b", ";;"]
    for _ in range(40):
        samples.append(" ".join(rng.choice(chunks) for _ in range(rng.randint(1, 12))))
    for code in samples:
        root = parse_c(code)  # must not raise
        sexps = all_subtree_sexps(root)
        assert isinstance(sexps, list)
        for sx in sexps[:50]:
            assert sx.count("(") == sx.count(")")  # well-formed s-exprs


def test_dfg_extractor_never_raises_fuzz():
    import random
    import string

    from deepdfa_amd.evaluator.dfg_c import get_data_flow, normalize_dataflow

    rng = random.Random(1)
    for _ in range(40):
        n = rng.randint(1, 100)
        code = "".join(rng.choice(string.printable) for _ in range(n))
        df = get_data_flow(code)  # must not raise
        normalize_dataflow(df)


def test_codebleu_degenerate_inputs():
    """calc_code_bleu must handle empty/degenerate candidates (models emit
    empty strings early in training) without raising or NaN."""
    from deepdfa_amd.evaluator.calc_code_bleu import calc_code_bleu

    ref = "int f(int a){return a+1;}"
    for cand in ["", ";", "(((", "int", ref]:
        out = calc_code_bleu([ref], [cand])
        score = out["code_bleu"]
        assert 0.0 <= score <= 1.0 and score == score  # finite, in range
        for k in ("ngram_match", "weighted_ngram_match", "syntax_match",
                  "dataflow_match"):
            assert k in out and out[k] == out[k]
    assert calc_code_bleu([ref], [ref])["code_bleu"] > 0.95
