"""CodeT5 generation-task driver (reference CodeT5/run_gen.py capability):
seq2seq fine-tuning (summarize / translate / refine / concode task shapes)
with per-epoch eval (smoothed BLEU) and beam-search generation at test.

Synthetic task (no-network env): target = the function's signature tokens
(a deterministic compressible mapping), mirroring the jsonl (src, tgt)
format of reference _utils.read_*_examples.
"""

from __future__ import annotations

import argparse
import logging
import os

import torch
from torch.utils.data import DataLoader, Dataset, RandomSampler

from ..data.text_dataset import synthetic_func_source
from ..data.tokenization import HashTokenizer
from ..evaluator import smoothed_bleu4
from ..models.t5 import T5Config, T5ForConditionalGeneration
from ..parallel.optim import FlatAdamW

logger = logging.getLogger(__name__)


class GenDataset(Dataset):
    """(source_ids, target_ids) pairs."""

    def __init__(self, tokenizer, n: int, max_source=128, max_target=32, seed=0):
        self.items = []
        for i in range(n):
            src = synthetic_func_source(seed * 10000 + i)
            tgt = " ".join(src.split("\n")[0].split()[:8])  # the signature
            s = tokenizer.encode(src, max_length=max_source)
            t = tokenizer.encode(tgt, max_length=max_target)
            self.items.append((torch.tensor(s), torch.tensor(t)))
        self.tokenizer = tokenizer

    def __len__(self):
        return len(self.items)

    def __getitem__(self, i):
        return self.items[i]


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--task", default="summarize",
                   choices=["summarize", "translate", "refine", "concode"])
    p.add_argument("--do_train", action="store_true")
    p.add_argument("--do_test", action="store_true")
    p.add_argument("--num_train_epochs", type=int, default=2)
    p.add_argument("--max_source_length", type=int, default=128)
    p.add_argument("--max_target_length", type=int, default=32)
    p.add_argument("--train_batch_size", type=int, default=8)
    p.add_argument("--learning_rate", type=float, default=5e-5)
    p.add_argument("--beam_size", type=int, default=2)
    p.add_argument("--n_synthetic", type=int, default=64)
    p.add_argument("--num_layers", type=int, default=2)
    p.add_argument("--d_model", type=int, default=128)
    p.add_argument("--output_dir", default="saved_models/gen")
    p.add_argument("--seed", type=int, default=0)
    args = p.parse_args(argv)
    logging.basicConfig(level=logging.INFO)
    torch.manual_seed(args.seed)
    device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    cfg = T5Config(num_layers=args.num_layers, num_decoder_layers=args.num_layers,
                   d_model=args.d_model, d_ff=args.d_model * 4,
                   num_heads=max(1, args.d_model // 64))
    tokenizer = HashTokenizer(vocab_size=cfg.vocab_size, cls=1, pad=0, sep=2)
    model = T5ForConditionalGeneration(cfg).to(device)
    train_ds = GenDataset(tokenizer, args.n_synthetic, args.max_source_length,
                          args.max_target_length, seed=1)
    test_ds = GenDataset(tokenizer, max(8, args.n_synthetic // 8),
                         args.max_source_length, args.max_target_length, seed=2)
    results = {}
    os.makedirs(args.output_dir, exist_ok=True)
    if args.do_train:
        opt = FlatAdamW(model.parameters(), lr=args.learning_rate)
        loader = DataLoader(train_ds, batch_size=args.train_batch_size,
                            sampler=RandomSampler(train_ds))
        model.train()
        for epoch in range(args.num_train_epochs):
            losses = []
            for src, tgt in loader:
                src, tgt = src.to(device), tgt.to(device)
                labels = tgt.masked_fill(tgt == cfg.pad_token_id, -100)
                loss, _logits, _ = model(src, labels=labels)
                opt.zero_grad()
                loss.backward()
                opt.clip_grad_norm_(1.0)
                opt.step()
                losses.append(float(loss.detach()))
            logger.info("epoch %d loss %.4f", epoch, sum(losses) / len(losses))
            results["train_loss"] = sum(losses) / len(losses)
        torch.save(model.state_dict(), os.path.join(args.output_dir, "pytorch_model.bin"))
    if args.do_test:
        model.eval()
        bleus = []
        for src, tgt in DataLoader(test_ds, batch_size=4):
            out = model.generate(src.to(device), max_length=args.max_target_length,
                                 num_beams=args.beam_size)
            for o, t in zip(out.cpu(), tgt):
                cand = [str(x) for x in o.tolist() if x not in (0, 1, 2)]
                ref = [str(x) for x in t.tolist() if x not in (0, 1, 2)]
                bleus.append(smoothed_bleu4(ref, cand))
        results["bleu4"] = sum(bleus) / max(1, len(bleus))
    return results


if __name__ == "__main__":
    main()
