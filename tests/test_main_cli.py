"""CLI parity tests: stacked configs, overrides, freeze_graph, analyzer."""

import os

import torch
import yaml

from deepdfa_amd.train import main_cli


def run(tmp_path, extra):
    root = str(tmp_path / "run")
    args = [
        "fit",
        "--config", "configs/config_default.yaml",
        "--config", "configs/config_bigvul.yaml",
        "--trainer.max_epochs", "1",
        "--trainer.default_root_dir", root,
        "--data.n_synthetic", "200",
        "--data.batch_size", "64",
    ] + extra
    return main_cli.main(args), root


def test_fit_with_stacked_configs(tmp_path, monkeypatch):
    monkeypatch.chdir(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    out, root = run(tmp_path, [])
    assert out["best_checkpoint"] is not None
    assert "best_validate" in out and "val_f1" in out["best_validate"]
    cfg = yaml.safe_load(open(os.path.join(root, "config.yaml")))
    assert cfg["data"]["batch_size"] == 64  # dotted override applied
    assert cfg["model"]["n_steps"] == 5
    assert os.path.exists(os.path.join(root, "output.log"))


def test_analyze_dataset(tmp_path, monkeypatch):
    monkeypatch.chdir(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    out, _ = run(tmp_path, ["--analyze_dataset"])
    assert "coverage" in out
    assert set(out["coverage"]) == {"train", "val", "test"}


def test_freeze_graph_transfer(tmp_path, monkeypatch):
    monkeypatch.chdir(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    out, root = run(tmp_path, [])
    ckpt = out["best_checkpoint"]
    # new run loading + freezing the graph weights
    from deepdfa_amd.models import FlowGNNGGNNModule

    torch.manual_seed(123)
    model = FlowGNNGGNNModule(input_dim=1002, hidden_dim=32, n_steps=5, num_output_layers=3)
    loaded = main_cli.freeze_graph_weights(model, ckpt)
    assert any(k.startswith("ggnn.") for k in loaded)
    frozen = [n for n, p in model.named_parameters() if not p.requires_grad]
    trainable = [n for n, p in model.named_parameters() if p.requires_grad]
    assert all(not n.startswith("output_layer") for n in frozen)
    assert all(n.startswith("output_layer") for n in trainable)
    # frozen weights equal checkpoint values
    payload = torch.load(ckpt, weights_only=False)
    assert torch.equal(
        model.state_dict()["ggnn.gru.weight_ih"], payload["state_dict"]["ggnn.gru.weight_ih"]
    )


def test_scalar_and_hpo_logging(tmp_path, monkeypatch):
    import json

    monkeypatch.chdir(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    out, root = run(tmp_path, [])
    scalars = [json.loads(l) for l in open(os.path.join(root, "scalars.jsonl"))]
    assert scalars and "train_loss" in scalars[0]
    hpo = [json.loads(l) for l in open(os.path.join(root, "hpo_metrics.jsonl"))]
    assert any("intermediate" in r for r in hpo)
    assert any("final" in r for r in hpo)


def test_fit_resume_through_cli(tmp_path, monkeypatch):
    """`fit --ckpt_path last.ckpt` resumes through main_cli: the resumed
    run trains only the remaining epochs (history length) and reuses the
    epoch counter from the checkpoint."""
    monkeypatch.chdir(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    out1, root1 = run(tmp_path, ["--trainer.max_epochs", "2",
                                 "--data.train_workers", "0"])
    last = None
    for dirpath, _dirs, files in os.walk(root1):
        if "last.ckpt" in files:
            last = os.path.join(dirpath, "last.ckpt")
    assert last, "last.ckpt not written"
    payload = torch.load(last, map_location="cpu", weights_only=False)
    assert payload["epoch"] == 1  # two epochs ran (0, 1)

    root2 = str(tmp_path / "resumed")
    out2 = main_cli.main([
        "fit",
        "--config", "configs/config_default.yaml",
        "--config", "configs/config_bigvul.yaml",
        "--trainer.max_epochs", "4",
        "--trainer.default_root_dir", root2,
        "--data.n_synthetic", "200",
        "--data.batch_size", "64",
        "--data.train_workers", "0",
        "--ckpt_path", last,
    ])
    assert len(out2["history"]) == 2  # epochs 2 and 3 only
    assert out2["history"][0]["epoch"] == 2
