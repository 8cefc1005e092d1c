"""End-to-end GPU training regression: a short LineVul fine-tune through
the production driver must show decreasing loss (catches integration bugs
like stale weight-cast caches that per-kernel numerics tests cannot)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.gpu
def test_linevul_short_training_loss_descends(tmp_path):
    from deepdfa_amd.data.text_dataset import TextDataset
    from deepdfa_amd.data.tokenization import HashTokenizer
    from deepdfa_amd.models.linevul import Model
    from deepdfa_amd.models.roberta import RobertaConfig

    torch.manual_seed(0)
    device = torch.device("cuda:0")
    cfg = RobertaConfig(num_hidden_layers=2)
    model = Model(config=cfg).to(device)
    tok = HashTokenizer(vocab_size=cfg.vocab_size)
    ds = TextDataset(tok, partition="train", block_size=128, n_synthetic=200)
    loader = torch.utils.data.DataLoader(ds, batch_size=16, shuffle=True,
                                         generator=torch.Generator().manual_seed(0))
    opt = torch.optim.AdamW(model.parameters(), lr=5e-5)
    losses = []
    model.train()
    for epoch in range(3):
        for ids, label, _ in loader:
            ids, label = ids.to(device), label.to(device)
            with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
                loss, _ = model(ids, labels=label)
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
            losses.append(float(loss.detach()))
    first = sum(losses[:5]) / 5
    last = sum(losses[-5:]) / 5
    assert last < first, (first, last)
    assert all(l == l for l in losses)  # no NaNs


@pytest.mark.gpu
def test_flat_adamw_training_loss_descends():
    """Same descent check through FlatAdamW (raw-kernel optimizer): guards
    the cast-cache epoch invalidation."""
    from deepdfa_amd.models.roberta import RobertaConfig, RobertaModel
    from deepdfa_amd.parallel.optim import FlatAdamW

    torch.manual_seed(0)
    device = torch.device("cuda:0")
    cfg = RobertaConfig(num_hidden_layers=2)
    enc = RobertaModel(cfg).to(device)
    head = torch.nn.Linear(cfg.hidden_size, 2).to(device)
    opt = FlatAdamW(list(enc.parameters()) + list(head.parameters()), lr=1e-4)
    ids = torch.randint(3, cfg.vocab_size, (16, 128), device=device)
    labels = torch.randint(0, 2, (16,), device=device)
    losses = []
    for _ in range(30):
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            hidden, _ = enc(ids)
            logits = head(hidden[:, 0].float())
        loss = torch.nn.functional.cross_entropy(logits, labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    # memorizing a fixed batch must drive the loss down hard; a stale
    # bf16 weight cache would keep it flat
    assert losses[-1] < 0.5 * losses[0], (losses[0], losses[-1])


@pytest.mark.gpu
def test_codet5_defect_training_loss_descends():
    """CodeT5 DefectModel short fine-tune on a fixed batch (T5 stack +
    relative-bias grads + fused pre-norm residuals)."""
    from deepdfa_amd.models.codet5 import DefectModel
    from deepdfa_amd.models.t5 import T5Config
    from deepdfa_amd.parallel.optim import FlatAdamW

    torch.manual_seed(0)
    device = torch.device("cuda:0")
    cfg = T5Config(num_layers=2, num_decoder_layers=2)
    model = DefectModel(config=cfg).to(device)
    opt = FlatAdamW(model.parameters(), lr=1e-4)
    ids = torch.randint(3, cfg.vocab_size, (8, 128), device=device)
    ids[:, -1] = cfg.eos_token_id
    labels = torch.randint(0, 2, (8,), device=device)
    losses = []
    for _ in range(25):
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            loss, _ = model(ids, labels=labels)
        opt.zero_grad()
        loss.backward()
        opt.clip_grad_norm_(1.0)
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < 0.7 * losses[0], (losses[0], losses[-1])


@pytest.mark.gpu
def test_trainer_fit_graph_capture_parity(tmp_path):
    """Production fit with hipGraph capture vs eager on the same seed/data:
    histories must match closely (capture pads batches with masked dummy
    graphs — numerically equivalent up to bf16 tiling differences)."""
    from deepdfa_amd.data.datamodule import BigVulDatasetLineVDDataModule
    from deepdfa_amd.models import FlowGNNGGNNModule
    from deepdfa_amd.parallel.optim import FlatAdamW
    from deepdfa_amd.train.trainer import Trainer

    results = {}
    for mode in ("eager", "capture"):
        torch.manual_seed(0)
        dm = BigVulDatasetLineVDDataModule(batch_size=32, n_synthetic=400,
                                           undersample="v1.0", seed=0)
        model = FlowGNNGGNNModule(input_dim=dm.input_dim, hidden_dim=32,
                                  n_steps=5, num_output_layers=3)
        trainer = Trainer(max_epochs=2, default_root_dir=str(tmp_path / mode),
                          precision="bf16", seed=0,
                          graph_capture=(mode == "capture"))
        opt = FlatAdamW(model.to(trainer.device).parameters(), lr=1e-3,
                        weight_decay=1e-2, l2_mode=True)
        out = trainer.fit(model, dm, optimizer=opt)
        results[mode] = out["history"]
    for r_e, r_c in zip(results["eager"], results["capture"]):
        assert abs(r_e["train_loss"] - r_c["train_loss"]) < 0.05, (r_e, r_c)
        # identical data order => identical counts up to bf16 threshold flips
        total = max(1.0, r_e["train_tp"] + r_e["train_fp"] + r_e["train_tn"] + r_e["train_fn"])
        assert r_e["train_tp"] + r_e["train_fp"] + r_e["train_tn"] + r_e["train_fn"] == \
               r_c["train_tp"] + r_c["train_fp"] + r_c["train_tn"] + r_c["train_fn"]
        assert abs(r_e["train_f1"] - r_c["train_f1"]) < 0.15, (r_e, r_c)
    assert abs(results["eager"][-1]["val_loss"] - results["capture"][-1]["val_loss"]) < 0.05


@pytest.mark.gpu
def test_direct_grad_accumulation_matches_autograd():
    """With FlatAdamW attached, weight-grad kernels accumulate DIRECTLY into
    the flat .grad views (ops/transformer.py backward direct paths) instead
    of returning tensors for AccumulateGrad. Same module, same input, grads
    must match the plain-autograd route across every op family: square
    linear (wgrad2 + colsum), fat linear, adjacent q/k/v (packed wgrad),
    rms_norm (atomic dgamma), embedding (scatter-add)."""
    import torch
    from torch import nn
    from deepdfa_amd.ops.transformer import (embedding_lookup, fused_linear,
                                             fused_qkv, rms_norm)
    from deepdfa_amd.parallel.optim import FlatAdamW

    torch.manual_seed(0)
    dev = "cuda"

    class Tiny(nn.Module):
        def __init__(self):
            super().__init__()
            self.emb = nn.Embedding(512, 768)
            self.gamma = nn.Parameter(torch.ones(768))
            self.wq = nn.Linear(768, 768, bias=False)
            self.wk = nn.Linear(768, 768, bias=False)
            self.wv = nn.Linear(768, 768, bias=False)
            self.o = nn.Linear(768, 768)
            self.wi = nn.Linear(768, 3072, bias=False)

        def forward(self, ids):
            e = embedding_lookup(ids, self.emb.weight)
            h = rms_norm(e, self.gamma).to(torch.bfloat16)
            qkv = fused_qkv(h, self.wq.weight, self.wk.weight, self.wv.weight)
            y = fused_linear(h, self.o.weight, self.o.bias)
            z = fused_linear(y, self.wi.weight, None)
            return qkv.float().square().mean() + z.float().square().mean()

    model = Tiny().to(dev)
    ids = torch.randint(0, 512, (4, 64), device=dev)

    loss_ref = model(ids)
    loss_ref.backward()
    ref = {n: p.grad.detach().clone() for n, p in model.named_parameters()}
    for p in model.parameters():
        p.grad = None

    opt = FlatAdamW(model.parameters(), lr=1e-3)
    opt.zero_grad()
    loss = model(ids)
    loss.backward()
    assert torch.allclose(loss.float(), loss_ref.float(), atol=1e-5)
    for n, p in model.named_parameters():
        r = ref[n].float()
        tol = 1e-4 + 1e-4 * r.abs().max()
        assert torch.allclose(p.grad.float(), r, atol=float(tol)), (
            n, (p.grad.float() - r).abs().max())


@pytest.mark.gpu
def test_ddfa_direct_grad_accumulation_matches_autograd():
    """FlowGNN direct-grad paths (embed4_direct, ggnn W_e/b_e, gate_pool,
    mlp3 — ops/flowgnn.py): grads with FlatAdamW attached must match the
    plain autograd route on the same batch."""
    import torch
    from deepdfa_amd.graph.synthetic import synthetic_cfg_batch
    from deepdfa_amd.models import FlowGNNGGNNModule
    from deepdfa_amd.parallel.optim import FlatAdamW

    torch.manual_seed(0)
    model = FlowGNNGGNNModule(input_dim=1002).to("cuda")
    g = synthetic_cfg_batch(64, seed=1).to("cuda")

    def step():
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            return model.training_step((g, {}))

    loss_ref = step()
    loss_ref.backward()
    ref = {n: p.grad.detach().clone() for n, p in model.named_parameters()}
    for p in model.parameters():
        p.grad = None
    model.metrics["train"].reset()

    opt = FlatAdamW(model.parameters(), lr=1e-3)
    opt.zero_grad()
    loss = step()
    loss.backward()
    assert torch.allclose(loss.float(), loss_ref.float(), atol=1e-4, rtol=1e-4)
    bad = []
    for n, p in model.named_parameters():
        r = ref[n].float()
        tol = 2e-4 + 2e-4 * float(r.abs().max())
        if not torch.allclose(p.grad.float(), r, atol=tol):
            bad.append((n, float((p.grad.float() - r).abs().max())))
    assert not bad, bad


@pytest.mark.gpu
def test_direct_grad_accumulation_sums_over_microbatches():
    """Gradient accumulation contract: with FlatAdamW, every direct-grad
    kernel must ACCUMULATE into .grad (never store) — two backwards before
    a step must yield exactly 2x the single-backward grads. Covers the
    mlp3 tail (+=), wgrad2 forced-atomic, gate_pool/embed4/ggnn atomics."""
    import torch
    from deepdfa_amd.graph.synthetic import synthetic_cfg_batch
    from deepdfa_amd.models import FlowGNNGGNNModule
    from deepdfa_amd.parallel.optim import FlatAdamW

    torch.manual_seed(0)
    model = FlowGNNGGNNModule(input_dim=1002).to("cuda")
    g = synthetic_cfg_batch(48, seed=7).to("cuda")
    opt = FlatAdamW(model.parameters(), lr=1e-3)

    def loss_fn():
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            return model.training_step((g, {}))

    opt.zero_grad()
    loss_fn().backward()
    once = opt.flat_g.detach().clone()
    opt.zero_grad()
    loss_fn().backward()
    loss_fn().backward()
    twice = opt.flat_g.detach().clone()
    err = (twice - 2 * once).abs().max() / once.abs().max().clamp(min=1e-8)
    assert float(err) < 1e-3, float(err)


@pytest.mark.gpu
def test_transformer_direct_grad_sums_over_microbatches():
    """Same accumulation contract for the transformer direct paths
    (wgrad2 forced-atomic, colsum, rmsnorm_wgrad, embed_scatter, packed
    q/k/v wgrad)."""
    import torch
    from deepdfa_amd.models.t5 import T5Config, T5ForConditionalGeneration
    from deepdfa_amd.parallel.optim import FlatAdamW

    torch.manual_seed(0)
    cfg = T5Config(num_layers=2, num_decoder_layers=2, vocab_size=1024,
                   dropout_rate=0.0)  # dropout redraws per microbatch
    model = T5ForConditionalGeneration(cfg).to("cuda")
    ids = torch.randint(3, cfg.vocab_size, (2, 64), device="cuda")
    opt = FlatAdamW(model.parameters(), lr=1e-3)

    def loss_fn():
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            dec = model(ids, labels=ids, output_hidden_only=True)
        return dec.float().square().mean()

    opt.zero_grad()
    loss_fn().backward()
    once = opt.flat_g.detach().clone()
    opt.zero_grad()
    loss_fn().backward()
    loss_fn().backward()
    twice = opt.flat_g.detach().clone()
    err = (twice - 2 * once).abs().max() / once.abs().max().clamp(min=1e-8)
    assert float(err) < 2e-3, float(err)
