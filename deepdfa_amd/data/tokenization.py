"""Tokenization for the transformer paths.

The reference uses pretrained HF tokenizers (microsoft/codebert-base BPE,
Salesforce/codet5-base; linevul_main.py:608-613 also supports a word-level
tokenizer). This environment has no network, so:

  * `load_pretrained_tokenizer(path)` loads any HF tokenizer directory when
    the user provides one (full parity path);
  * `train_bpe_tokenizer(corpus)` trains a byte-level BPE offline with the
    `tokenizers` library (the reference's word_level/bpe tokenizer-training
    capability, LineVul/linevul/*_tokenizer/ + CodeT5/tokenizer/);
  * `HashTokenizer` is the deterministic no-asset fallback used by synthetic
    benchmarks: splits code into sub-tokens (IVDetect-style camelCase +
    punctuation split, sastvd/helpers/tokenise.py) and hashes them into the
    vocabulary.

All three expose the small surface the drivers use: cls/sep/pad ids and
encode(text, max_length) -> List[int].
"""

from __future__ import annotations

import re
from typing import List

from .. import hashstr


def tokenise(s: str) -> List[str]:
    """IVDetect-style subtoken split (reference sastvd/helpers/tokenise.py:4-35):
    split on non-alphanumeric, then split camelCase and letter/digit
    boundaries."""
    out: List[str] = []
    for tok in re.split(r"[^A-Za-z0-9]+", s):
        if not tok:
            continue
        parts = re.findall(r"[A-Z]+(?=[A-Z][a-z])|[A-Z]?[a-z]+|[A-Z]+|\d+", tok)
        out.extend(p for p in parts if p)
    return out


def tokenise_lines(s: str) -> List[List[str]]:
    """Per-line subtoken split (reference tokenise.py:23-35)."""
    return [tokenise(line) for line in s.splitlines()]


class HashTokenizer:
    """Deterministic vocabulary-hashing tokenizer (no assets needed)."""

    def __init__(self, vocab_size: int = 50265, cls=0, pad=1, sep=2, unk=3, reserved=4):
        self.vocab_size = vocab_size
        self.cls_token_id = cls
        self.pad_token_id = pad
        self.sep_token_id = sep
        self.unk_token_id = unk
        self._reserved = reserved

    def _tok2id(self, tok: str) -> int:
        return self._reserved + hashstr("tok:" + tok) % (self.vocab_size - self._reserved)

    def encode(self, text: str, max_length: int = 512) -> List[int]:
        ids = [self.cls_token_id]
        for tok in tokenise(text)[: max_length - 2]:
            ids.append(self._tok2id(tok))
        ids.append(self.sep_token_id)
        ids = ids[:max_length]
        ids += [self.pad_token_id] * (max_length - len(ids))
        return ids

    def __len__(self):
        return self.vocab_size


def synthetic_corpus(n: int = 256, seed: int = 0) -> List[str]:
    """Synthetic C-function corpus for offline tokenizer training."""
    from .text_dataset import synthetic_func_source

    return [synthetic_func_source(seed * 100000 + i) for i in range(n)]


def train_bpe_tokenizer(corpus: List[str], vocab_size: int = 50265):
    """Train a byte-level BPE offline (HF `tokenizers`)."""
    from tokenizers import Tokenizer, models, pre_tokenizers, trainers

    tok = Tokenizer(models.BPE(unk_token="<unk>"))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    trainer = trainers.BpeTrainer(
        vocab_size=vocab_size, special_tokens=["<s>", "<pad>", "</s>", "<unk>", "<mask>"]
    )
    tok.train_from_iterator(corpus, trainer)
    return tok


def train_word_level_tokenizer(corpus: List[str], vocab_size: int = 50265):
    """Train a whitespace word-level tokenizer offline (the reference's
    --use_word_level_tokenizer path, linevul_main.py:608-610, which loads a
    pretrained word_level/train_word_level.json asset)."""
    from tokenizers import Tokenizer, models, pre_tokenizers, trainers

    tok = Tokenizer(models.WordLevel(unk_token="<unk>"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()
    trainer = trainers.WordLevelTrainer(
        vocab_size=vocab_size, special_tokens=["<s>", "<pad>", "</s>", "<unk>", "<mask>"]
    )
    tok.train_from_iterator(corpus, trainer)
    return tok


def load_pretrained_tokenizer(path: str):
    from transformers import AutoTokenizer

    return AutoTokenizer.from_pretrained(path)
