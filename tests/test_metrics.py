"""BinaryStats / pr_curve / classification_report vs sklearn references
(the reference uses torchmetrics + sklearn.classification_report,
base_module.py:34-68, 325-383 — sklearn here is the independent oracle)."""

import torch

from deepdfa_amd.utils.metrics import (BinaryStats, classification_report_dict,
                                       pr_curve)


def _rand(n=500, seed=0):
    g = torch.Generator().manual_seed(seed)
    probs = torch.rand(n, generator=g)
    labels = (torch.rand(n, generator=g) < 0.3).long()
    return probs, labels


def test_binary_stats_vs_sklearn():
    from sklearn.metrics import (accuracy_score, f1_score, precision_score,
                                 recall_score)

    probs, labels = _rand()
    st = BinaryStats(0.5)
    # incremental updates must accumulate
    st.update(probs[:200], labels[:200])
    st.update(probs[200:], labels[200:])
    out = st.compute()
    pred = (probs >= 0.5).long().numpy()
    lab = labels.numpy()
    assert abs(out["accuracy"] - accuracy_score(lab, pred)) < 1e-9
    assert abs(out["precision"] - precision_score(lab, pred)) < 1e-9
    assert abs(out["recall"] - recall_score(lab, pred)) < 1e-9
    assert abs(out["f1"] - f1_score(lab, pred)) < 1e-9


def test_binary_stats_mask_equals_subset():
    probs, labels = _rand(seed=3)
    mask = torch.rand(500) < 0.7
    a = BinaryStats(0.5)
    a.update(probs, labels, mask=mask)
    b = BinaryStats(0.5)
    b.update(probs[mask], labels[mask])
    assert torch.equal(a.counts, b.counts)


def test_classification_report_vs_sklearn():
    from sklearn.metrics import classification_report

    probs, labels = _rand(seed=7)
    ours = classification_report_dict(probs, labels)
    ref = classification_report(labels.numpy(), (probs >= 0.5).long().numpy(),
                                output_dict=True)
    for cls in ("0", "1"):
        for k in ("precision", "recall", "f1-score", "support"):
            assert abs(float(ours[cls][k]) - float(ref[cls][k])) < 1e-9, (cls, k)


def test_pr_curve_monotone_recall_and_endpoints():
    from sklearn.metrics import precision_score, recall_score

    probs, labels = _rand(seed=11)
    ths, precs, recs = pr_curve(probs, labels, num_thresholds=51)
    assert len(ths) == len(precs) == len(recs) == 51
    # recall is non-increasing as the threshold rises; endpoints sane
    assert all(recs[i] >= recs[i + 1] - 1e-12 for i in range(len(recs) - 1))
    assert recs[0] == 1.0  # threshold 0 predicts everything positive
    # spot-check one interior threshold against sklearn
    t = ths[25]
    pred = (probs >= t).long().numpy()
    assert abs(precs[25] - precision_score(labels.numpy(), pred)) < 1e-9
    assert abs(recs[25] - recall_score(labels.numpy(), pred)) < 1e-9
