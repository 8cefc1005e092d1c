"""TextDataset for the LineVul / CodeT5 drivers.

Parity target: reference LineVul/linevul/linevul_main.py:55-131
(TextDataset + convert_examples_to_features: tokenize each function to
exactly block_size=512 ids with CLS/SEP and PAD fill, keep (input_ids,
label, index)) and CodeT5/utils.py:76-101 (load_and_cache_defect_data:
TensorDataset of (source_ids, label, idx) from jsonl with caching).

Sources: a pandas dataframe with columns (id, func, vul) loaded from CSV /
jsonl, or the synthetic Big-Vul-shaped generator (no-network benchmarks).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import numpy as np
import pandas as pd
import torch
from torch.utils.data import Dataset

from .dclass import synthetic_bigvul_df


@dataclass
class InputFeatures:
    input_tokens: List[str]
    input_ids: List[int]
    label: int
    index: int


def synthetic_func_source(_id: int, n_lines: int = 12, vul: int = 0) -> str:
    """Deterministic pseudo-C function for tokenizer-driven paths. When
    `vul` is set the function contains an unchecked strcpy/alloca pattern —
    a LEARNABLE signal, so end-to-end training demos and regression tests
    show real F1 movement instead of fitting noise (real Big-Vul data
    plugs in via the CSV/jsonl loaders)."""
    rng = np.random.RandomState(_id % (2**31))
    names = ["buf", "len", "ptr", "data", "size", "idx", "tmp", "out", "src", "dst"]
    types = ["int", "char *", "size_t", "uint32_t", "void *"]
    lines = [f"static int func_{_id}({types[rng.randint(5)]} {names[rng.randint(10)]}) {{"]
    for i in range(n_lines):
        a, b, c = (names[rng.randint(10)] for _ in range(3))
        op = ["+", "-", "*", "="][rng.randint(4)]
        if rng.rand() < 0.3:
            lines.append(f"  if ({a} {op} {b} > {rng.randint(1024)}) return -1;")
        elif rng.rand() < 0.4:
            lines.append(f"  memcpy({a}, {b}, {c});")
        else:
            lines.append(f"  {a} = {b} {op} {c};")
    if vul:
        pos = 1 + rng.randint(max(1, len(lines) - 1))
        lines.insert(pos, f"  strcpy({names[rng.randint(10)]}, user_input);")
    lines.append("  return 0;\n}")
    return "\n".join(lines)


def convert_examples_to_features(func: str, label: int, index: int, tokenizer, block_size=512):
    """linevul_main.py:110-131 contract: truncate to block_size-2 sub-tokens,
    wrap with CLS/SEP, pad to block_size."""
    if hasattr(tokenizer, "encode") and not hasattr(tokenizer, "tokenize"):
        ids = tokenizer.encode(func, max_length=block_size)
        toks = []
    else:  # HF tokenizer
        toks = tokenizer.tokenize(str(func))[: block_size - 2]
        toks = [tokenizer.cls_token] + toks + [tokenizer.sep_token]
        ids = tokenizer.convert_tokens_to_ids(toks)
        ids += [tokenizer.pad_token_id] * (block_size - len(ids))
    return InputFeatures(toks, ids, int(label), int(index))


class TextDataset(Dataset):
    def __init__(
        self,
        tokenizer,
        args=None,
        file_path: Optional[str] = None,
        df: Optional[pd.DataFrame] = None,
        partition: str = "train",
        block_size: int = 512,
        n_synthetic: int = 2000,
        split: str = "fixed",
    ):
        self.tokenizer = tokenizer
        self.block_size = block_size
        if df is None:
            if file_path is not None:
                if file_path.endswith(".jsonl"):
                    df = pd.read_json(file_path, lines=True)
                    df = df.rename(columns={"target": "vul", "code": "func"})
                    if "id" not in df.columns:
                        df = df.rename(columns={"idx": "id"})
                else:
                    df = pd.read_csv(file_path)
                    if "processed_func" in df.columns:
                        df = df.rename(columns={"processed_func": "func", "target": "vul"})
            else:
                from .dclass import ds_partition

                df = synthetic_bigvul_df(n_synthetic)
                df = ds_partition(df, partition, split=split)
                df = df.assign(
                    func=[synthetic_func_source(i, vul=v) for i, v in zip(df.id, df.vul)]
                )
        self.df = df.reset_index(drop=True)
        self.examples = [
            convert_examples_to_features(
                row.func, row.vul, row.id, tokenizer, block_size=block_size
            )
            for row in self.df.itertuples()
        ]

    def __len__(self):
        return len(self.examples)

    def __getitem__(self, i):
        ex = self.examples[i]
        return (
            torch.tensor(ex.input_ids, dtype=torch.long),
            torch.tensor(ex.label, dtype=torch.long),
            torch.tensor(ex.index, dtype=torch.long),
        )
