"""Multi-task generation driver (reference CodeT5/run_multi_gen.py
capability): one shared seq2seq model trained across several generation
tasks with size-proportional (^0.7 temperature-smoothed) task sampling,
cycled per-task dataloaders, per-task patience early-stopping, and
per-task BLEU eval — the CodeT5 multi-task fine-tuning recipe
(run_multi_gen.py:269-288 sampling, :248-267 patience table,
:86-175 eval_bleu).

Synthetic tasks in this no-network env: each "task" maps source code to a
different deterministic target (signature / identifiers / reversed
tokens), exercising the same multi-task machinery.
"""

from __future__ import annotations

import argparse
import logging
import os
from itertools import cycle

import numpy as np
import torch
from torch.utils.data import DataLoader, RandomSampler

from ..data.text_dataset import synthetic_func_source
from ..data.tokenization import HashTokenizer
from ..evaluator import smoothed_bleu4
from ..models.t5 import T5Config, T5ForConditionalGeneration
from ..parallel.optim import FlatAdamW
from .run_gen import GenDataset

logger = logging.getLogger(__name__)

# reference patience table (run_multi_gen.py:254-266)
PATIENCE = {"summarize": 2, "translate": 5, "refine": 5, "concode": 3, "defect": 2}

MAX_TARGET_LEN = {"summarize": 128, "translate": 256, "refine": 240,
                  "concode": 150, "defect": 3}


def _target_fn(task: str):
    if task == "summarize":
        return lambda src: " ".join(src.split("\n")[0].split()[:8])
    if task == "translate":
        return lambda src: " ".join(reversed(src.split()[:16]))
    if task == "refine":
        return lambda src: src.replace("int ", "long ")[:120]
    return lambda src: " ".join(sorted(set(src.split()))[:10])


class MultiGenDataset(GenDataset):
    def __init__(self, tokenizer, task: str, n: int, max_source=128, max_target=32, seed=0):
        self.items = []
        fn = _target_fn(task)
        for i in range(n):
            src = synthetic_func_source(seed * 10000 + i)
            s = tokenizer.encode(src, max_length=max_source)
            t = tokenizer.encode(fn(src), max_length=max_target)
            self.items.append((torch.tensor(s), torch.tensor(t)))
        self.tokenizer = tokenizer


def sampling_probs(sizes):
    """Size-proportional with ^0.7 temperature (run_multi_gen.py:269-272)."""
    p = np.asarray(sizes, dtype=np.float64)
    p = p / p.sum()
    p = p ** 0.7
    return p / p.sum()


def eval_bleu(model, loader, device, beam_size, max_target):
    model.eval()
    bleus = []
    with torch.no_grad():
        for src, tgt in loader:
            out = model.generate(src.to(device), max_length=max_target, num_beams=beam_size)
            for o, t in zip(out.cpu(), tgt):
                cand = [str(x) for x in o.tolist() if x not in (0, 1, 2)]
                ref = [str(x) for x in t.tolist() if x not in (0, 1, 2)]
                bleus.append(smoothed_bleu4(ref, cand))
    model.train()
    return sum(bleus) / max(1, len(bleus))


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--tasks", default="summarize,translate",
                   help="comma-separated task list")
    p.add_argument("--max_steps", type=int, default=40)
    p.add_argument("--eval_every", type=int, default=20)
    p.add_argument("--train_batch_size", type=int, default=8)
    p.add_argument("--max_source_length", type=int, default=128)
    p.add_argument("--max_target_length", type=int, default=32)
    p.add_argument("--learning_rate", type=float, default=5e-5)
    p.add_argument("--beam_size", type=int, default=2)
    p.add_argument("--n_synthetic", type=int, default=48)
    p.add_argument("--num_layers", type=int, default=2)
    p.add_argument("--d_model", type=int, default=128)
    p.add_argument("--output_dir", default="saved_models/multi_gen")
    p.add_argument("--seed", type=int, default=0)
    args = p.parse_args(argv)
    logging.basicConfig(level=logging.INFO)
    torch.manual_seed(args.seed)
    rng = np.random.RandomState(args.seed)
    device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    tasks = args.tasks.split(",")
    cfg = T5Config(num_layers=args.num_layers, num_decoder_layers=args.num_layers,
                   d_model=args.d_model, d_ff=args.d_model * 4,
                   num_heads=max(1, args.d_model // 64))
    tokenizer = HashTokenizer(vocab_size=cfg.vocab_size, cls=1, pad=0, sep=2)
    model = T5ForConditionalGeneration(cfg).to(device)
    opt = FlatAdamW(model.parameters(), lr=args.learning_rate)

    # sized differently per task so the ^0.7 sampling is non-trivial
    train = {
        t: MultiGenDataset(tokenizer, t, args.n_synthetic * (i + 1),
                           args.max_source_length, args.max_target_length, seed=1 + i)
        for i, t in enumerate(tasks)
    }
    dev = {
        t: MultiGenDataset(tokenizer, t, 8, args.max_source_length,
                           args.max_target_length, seed=100 + i)
        for i, t in enumerate(tasks)
    }
    loaders = {
        t: cycle(DataLoader(d, batch_size=args.train_batch_size, sampler=RandomSampler(d)))
        for t, d in train.items()
    }
    probs = sampling_probs([len(d) for d in train.values()])
    best_bleu = {t: -1.0 for t in tasks}
    stale = {t: 0 for t in tasks}
    stopped = {t: False for t in tasks}
    os.makedirs(args.output_dir, exist_ok=True)
    model.train()
    step, skip = 0, 0
    losses = {t: [] for t in tasks}
    while step < args.max_steps:
        task = str(rng.choice(tasks, p=probs))
        if stopped[task]:
            skip += 1
            if skip > 50:
                logger.info("all tasks early-stopped at step %d", step)
                break
            continue
        skip = 0
        step += 1
        src, tgt = next(loaders[task])
        src, tgt = src.to(device), tgt.to(device)
        labels = tgt.masked_fill(tgt == cfg.pad_token_id, -100)
        loss, _logits, _ = model(src, labels=labels)
        opt.zero_grad()
        loss.backward()
        opt.clip_grad_norm_(1.0)
        opt.step()
        losses[task].append(float(loss.detach()))
        if step % args.eval_every == 0:
            for t in tasks:
                if stopped[t]:
                    continue
                bleu = eval_bleu(model, DataLoader(dev[t], batch_size=4), device,
                                 args.beam_size, args.max_target_length)
                if bleu > best_bleu[t]:
                    best_bleu[t] = bleu
                    stale[t] = 0
                    d = os.path.join(args.output_dir, f"checkpoint-best-bleu-{t}")
                    os.makedirs(d, exist_ok=True)
                    torch.save(model.state_dict(), os.path.join(d, "pytorch_model.bin"))
                else:
                    stale[t] += 1
                    if stale[t] >= PATIENCE.get(t, 3):
                        stopped[t] = True
                        logger.info("task %s early-stopped (patience %d)", t, stale[t])
                logger.info("step %d task %s bleu %.2f best %.2f", step, t, bleu, best_bleu[t])
    torch.save(model.state_dict(),
               os.path.join(args.output_dir, "checkpoint-last.bin"))
    return {
        "steps": step,
        "best_bleu": best_bleu,
        "train_loss": {t: (sum(v) / len(v) if v else None) for t, v in losses.items()},
    }


if __name__ == "__main__":
    main()
