"""Trainer loop tests: fit/val/test, checkpoint naming + best selection."""

import glob
import os

import torch

from deepdfa_amd.data import BigVulDatasetLineVDDataModule
from deepdfa_amd.models import FlowGNNGGNNModule
from deepdfa_amd.train import Trainer

FEAT = "_ABS_DATAFLOW_datatype_all_limitall_1000_limitsubkeys_1000"


def small_setup(tmp_path, epochs=2):
    dm = BigVulDatasetLineVDDataModule(
        feat=FEAT, batch_size=64, n_synthetic=300, undersample="v1.0"
    )
    torch.manual_seed(0)
    model = FlowGNNGGNNModule(
        feat=FEAT, input_dim=dm.input_dim, hidden_dim=32, n_steps=5, num_output_layers=3
    )
    tr = Trainer(max_epochs=epochs, default_root_dir=str(tmp_path), periodic_every=2)
    return dm, model, tr


def test_fit_creates_checkpoints(tmp_path):
    dm, model, tr = small_setup(tmp_path)
    out = tr.fit(model, dm)
    assert len(out["history"]) == 2
    ckpts = os.listdir(tr.ckpt_dir)
    assert "last.ckpt" in ckpts
    assert any(c.startswith("performance-") for c in ckpts)
    assert any(c.startswith("periodical-") for c in ckpts)
    best = out["best_checkpoint"]
    assert best is not None and os.path.exists(best)
    # only one performance ckpt kept (save_top_k=1)
    assert len(glob.glob(os.path.join(tr.ckpt_dir, "performance-*.ckpt"))) == 1


def test_checkpoint_roundtrip(tmp_path):
    dm, model, tr = small_setup(tmp_path, epochs=1)
    out = tr.fit(model, dm)
    model2 = FlowGNNGGNNModule(
        feat=FEAT, input_dim=dm.input_dim, hidden_dim=32, n_steps=5, num_output_layers=3
    )
    payload = tr.load_checkpoint(model2, os.path.join(tr.ckpt_dir, "last.ckpt"))
    assert payload["epoch"] == 0
    for (n1, p1), (n2, p2) in zip(model.named_parameters(), model2.named_parameters()):
        assert n1 == n2
        assert torch.equal(p1, p2)


def test_test_loop_metrics(tmp_path):
    dm, model, tr = small_setup(tmp_path, epochs=1)
    tr.fit(model, dm)
    res = tr.test(model, dm)
    assert "test_f1" in res and "test_1_recall" in res
    assert os.path.exists(os.path.join(str(tmp_path), "pr.csv"))
    assert "classification_report" in res


def test_loss_decreases(tmp_path):
    dm, model, tr = small_setup(tmp_path, epochs=4)
    out = tr.fit(model, dm)
    losses = [h["train_loss"] for h in out["history"]]
    assert losses[-1] < losses[0]


def test_fit_resume_matches_uninterrupted(tmp_path):
    """fit(ckpt_path=last.ckpt) resumes weights + optimizer + epoch/step
    counters and replays the completed epochs' loader draws: 2 epochs +
    resume for 2 more == 4 uninterrupted epochs (identical final params)."""
    import torch

    from deepdfa_amd.data.datamodule import BigVulDatasetLineVDDataModule
    from deepdfa_amd.models import FlowGNNGGNNModule
    from deepdfa_amd.train.trainer import Trainer

    def make():
        torch.manual_seed(0)
        dm = BigVulDatasetLineVDDataModule(batch_size=8, n_synthetic=200,
                                           undersample="v1.0", seed=0)
        model = FlowGNNGGNNModule(input_dim=1002, hidden_dim=8, n_steps=2,
                                  num_output_layers=2)
        opt = torch.optim.Adam(model.parameters(), lr=1e-3)
        return dm, model, opt

    dm, model, opt = make()
    tr = Trainer(max_epochs=4, default_root_dir=str(tmp_path / "full"), seed=3)
    tr.fit(model, dm, optimizer=opt)
    ref = torch.cat([p.detach().flatten() for p in model.parameters()])

    dm2, model2, opt2 = make()
    tr2 = Trainer(max_epochs=2, default_root_dir=str(tmp_path / "half"), seed=3)
    tr2.fit(model2, dm2, optimizer=opt2)
    last = os.path.join(tr2.ckpt_dir, "last.ckpt")
    assert os.path.exists(last)

    dm3, model3, opt3 = make()
    tr3 = Trainer(max_epochs=4, default_root_dir=str(tmp_path / "resume"), seed=3)
    out = tr3.fit(model3, dm3, optimizer=opt3, ckpt_path=last)
    got = torch.cat([p.detach().flatten() for p in model3.parameters()])
    assert torch.allclose(got, ref, atol=1e-6), float((got - ref).abs().max())
    assert len(out["history"]) == 2  # only the resumed epochs ran


def test_resume_same_dir_keeps_historical_best(tmp_path):
    """Resuming into the SAME run dir restores the best-val watermark from
    the existing performance-*.ckpt names: a worse post-resume epoch must
    not displace the historical best checkpoint."""
    import glob

    import torch

    from deepdfa_amd.data.datamodule import BigVulDatasetLineVDDataModule
    from deepdfa_amd.models import FlowGNNGGNNModule
    from deepdfa_amd.train.trainer import Trainer

    torch.manual_seed(0)
    dm = BigVulDatasetLineVDDataModule(batch_size=8, n_synthetic=200,
                                       undersample="v1.0", seed=0)
    model = FlowGNNGGNNModule(input_dim=1002, hidden_dim=8, n_steps=2,
                              num_output_layers=2)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    root = str(tmp_path / "run")
    tr = Trainer(max_epochs=2, default_root_dir=root, seed=3)
    tr.fit(model, dm, optimizer=opt)
    best_before = tr.best_checkpoint()
    assert best_before

    tr2 = Trainer(max_epochs=4, default_root_dir=root, seed=3)
    tr2.ckpt_dir = tr.ckpt_dir  # same run dir (fresh Trainer objects stamp new dirs)
    tr2.fit(model, dm, optimizer=opt,
            ckpt_path=os.path.join(tr.ckpt_dir, "last.ckpt"))
    # watermark restored: at most one best file exists per val-loss record,
    # and the selected best is never worse than the pre-resume best
    import re as _re

    def loss_of(p):
        return float(_re.search(r"performance-\d+-\d+-([\d.]+)\.ckpt$", p).group(1))

    best_after = tr2.best_checkpoint()
    assert loss_of(best_after) <= loss_of(best_before) + 1e-9


def test_accumulate_grad_batches_matches_joint_step(tmp_path):
    """Lightning-parity accumulate_grad_batches: two equal micro-batches at
    accum=2 produce EXACTLY the same update as one step on the averaged
    loss (loss/accum per micro-batch, zero at group start, step at end)."""
    import copy

    import torch

    from deepdfa_amd.graph.synthetic import synthetic_cfg_batch
    from deepdfa_amd.models import FlowGNNGGNNModule
    from deepdfa_amd.train.trainer import Trainer

    torch.manual_seed(0)
    model = FlowGNNGGNNModule(input_dim=1002, hidden_dim=8, n_steps=2,
                              num_output_layers=2)
    ref = copy.deepcopy(model)
    b1 = synthetic_cfg_batch(8, seed=1)
    b2 = synthetic_cfg_batch(8, seed=2)

    class StubDM:
        batch_size = 8

        def train_dataloader(self, generator=None, rank=0, world=1):
            return [(b1, {}), (b2, {})]

        def val_dataloader(self, rank=0, world=1):
            return []

    tr = Trainer(max_epochs=1, default_root_dir=str(tmp_path), seed=0,
                 precision="fp32", accumulate_grad_batches=2,
                 check_val_every_n_epoch=100, periodic_every=100)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    tr.fit(model, StubDM(), optimizer=opt)

    # manual reference: one step on the mean of the two batch losses
    opt_ref = torch.optim.SGD(ref.parameters(), lr=0.1)
    opt_ref.zero_grad()
    loss = (ref.training_step((b1, {})) + ref.training_step((b2, {}))) / 2
    loss.backward()
    opt_ref.step()

    for (n, p), (_, q) in zip(model.named_parameters(), ref.named_parameters()):
        assert torch.allclose(p, q, atol=1e-7), n
