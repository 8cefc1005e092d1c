"""deepdfa_amd — MI355X-native vulnerability-detection training framework.

A from-scratch rebuild of the capabilities of the ISU-PAAL/DeepDFA artifact
(reference: /root/reference) designed MI355X-first:

  * batched CFGs as block-diagonal CSR/CSC (no DGL),
  * hand-written HIP/CDNA4 (gfx950) kernels for the flow-GNN message passing,
    attention pooling, fused GRU, transformer blocks, losses and optimizers,
  * RCCL over xGMI for data-parallel training (one process per GPU),
  * hipGraph-captured training steps for the launch-bound small-model regime.

Layer map mirrors the reference:
  graph/     batched-CFG substrate        (replaces DGL graphs + graphmogrifier)
  data/      datasets, splits, features   (replaces sastvd/helpers + linevd)
  ops/       HIP kernels + CPU references (replaces DGL/cuDNN delegated kernels)
  models/    FlowGNN / RoBERTa / T5 / combined models
  parallel/  RCCL DDP engine, rank-consistent sampling
  train/     trainer, CLI, checkpointing
  utils/     logging, metrics, profiling
"""

import hashlib
import os
from multiprocessing import Pool

__version__ = "0.1.0"

# ---------------------------------------------------------------------------
# Storage layout (mirrors reference DDFA/sastvd/__init__.py:42-120 semantics:
# a storage root with external/processed/cache/interim/outputs subdirs,
# overridable via env var).
# ---------------------------------------------------------------------------

def storage_root() -> str:
    return os.environ.get(
        "DEEPDFA_STORAGE", os.path.join(os.path.dirname(os.path.dirname(__file__)), "storage")
    )


def _subdir(name: str, *paths: str) -> str:
    p = os.path.join(storage_root(), name, *paths)
    os.makedirs(os.path.dirname(p) if os.path.splitext(p)[1] else p, exist_ok=True)
    return p


def external_dir(*paths: str) -> str:
    return _subdir("external", *paths)


def processed_dir(*paths: str) -> str:
    return _subdir("processed", *paths)


def cache_dir(*paths: str) -> str:
    return _subdir("cache", *paths)


def interim_dir(*paths: str) -> str:
    return _subdir("interim", *paths)


def outputs_dir(*paths: str) -> str:
    return _subdir("outputs", *paths)


def hashstr(s) -> int:
    """Deterministic string hash (reference sastvd/__init__.py:190-196 contract:
    stable across processes, unlike Python's salted hash())."""
    return int(hashlib.sha1(str(s).encode("utf-8")).hexdigest(), 16) % (10**8)


def dfmp(df, function, columns=None, ordr=True, workers=6, cs=10, desc=""):
    """Parallel map over dataframe rows (reference sastvd/__init__.py:198-244
    contract: map `function` over rows (or a single column) with a process
    pool, preserving order when ordr=True)."""
    if columns is None:
        items = [row for _, row in df.iterrows()]
    elif isinstance(columns, str):
        items = df[columns].tolist()
    else:
        items = [dict(row[columns]) for _, row in df.iterrows()]
    if workers <= 1:
        return [function(i) for i in items]
    with Pool(processes=workers) as pool:
        mapper = pool.imap if ordr else pool.imap_unordered
        return list(mapper(function, items, cs))
