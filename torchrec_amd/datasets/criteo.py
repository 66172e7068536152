"""Criteo dataset loaders.

Reference parity: torchrec/datasets/criteo.py (criteo_terabyte /
criteo_kaggle :143,171 TSV pipes, BinaryCriteoUtils :198 npy conversion,
InMemoryBinaryCriteoIterDataPipe :715 with per-rank sharding).
"""

from __future__ import annotations

import os
from typing import Any, Dict, Iterator, List, Optional, Tuple

import numpy as np
import torch
from torch.utils.data import IterableDataset

from torchrec_amd.datasets.random import Batch
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor

INT_FEATURE_COUNT = 13
CAT_FEATURE_COUNT = 26
DAYS = 24
DEFAULT_LABEL_NAME = "label"
DEFAULT_INT_NAMES: List[str] = [f"int_{i}" for i in range(INT_FEATURE_COUNT)]
DEFAULT_CAT_NAMES: List[str] = [f"cat_{i}" for i in range(CAT_FEATURE_COUNT)]
DEFAULT_COLUMN_NAMES: List[str] = [DEFAULT_LABEL_NAME, *DEFAULT_INT_NAMES, *DEFAULT_CAT_NAMES]


def _parse_tsv_line(line: str) -> Tuple[int, List[int], List[int]]:
    cols = line.rstrip("\n").split("\t")
    label = int(cols[0]) if cols[0] else 0
    dense = [
        int(c) if c else 0 for c in cols[1 : 1 + INT_FEATURE_COUNT]
    ]
    sparse = [
        int(c, 16) if c else 0
        for c in cols[1 + INT_FEATURE_COUNT : 1 + INT_FEATURE_COUNT + CAT_FEATURE_COUNT]
    ]
    return label, dense, sparse


class CriteoIterDataPipe(IterableDataset):
    """Row-wise TSV reader (reference criteo.py:55 _default_row_mapper path)."""

    def __init__(self, paths: List[str], row_mapper=None) -> None:
        self.paths = paths
        self.row_mapper = row_mapper

    def __iter__(self) -> Iterator[Any]:
        for path in self.paths:
            opener = open
            if path.endswith(".gz"):
                import gzip

                opener = gzip.open
            with opener(path, "rt") as f:
                for line in f:
                    row = _parse_tsv_line(line)
                    yield self.row_mapper(row) if self.row_mapper else row


def criteo_terabyte(paths: List[str], row_mapper=None) -> CriteoIterDataPipe:
    """Reference parity: datasets/criteo.py:143."""
    return CriteoIterDataPipe(list(paths), row_mapper)


def criteo_kaggle(path: str, row_mapper=None) -> CriteoIterDataPipe:
    """Reference parity: datasets/criteo.py:171."""
    return CriteoIterDataPipe([path], row_mapper)


class BinaryCriteoUtils:
    """TSV -> dense/sparse/labels .npy conversion + shard math
    (reference criteo.py:198)."""

    @staticmethod
    def tsv_to_npys(in_path: str, out_dense: str, out_sparse: str, out_labels: str) -> None:
        labels, dense, sparse = [], [], []
        for label, d, s in CriteoIterDataPipe([in_path]):
            labels.append(label)
            dense.append(d)
            sparse.append(s)
        np.save(out_dense, np.array(dense, dtype=np.float32))
        np.save(out_sparse, np.array(sparse, dtype=np.int64))
        np.save(out_labels, np.array(labels, dtype=np.int32).reshape(-1, 1))

    @staticmethod
    def get_shape_from_npy(path: str) -> Tuple[int, ...]:
        with open(path, "rb") as f:
            np.lib.format.read_magic(f)
            shape, _, _ = np.lib.format.read_array_header_1_0(f)
        return shape

    @staticmethod
    def shuffle_npys(dense: np.ndarray, sparse: np.ndarray, labels: np.ndarray, seed: int = 0):
        rng = np.random.default_rng(seed)
        perm = rng.permutation(dense.shape[0])
        return dense[perm], sparse[perm], labels[perm]

    @staticmethod
    def get_file_row_ranges_and_remainder(
        lengths: List[int], rank: int, world_size: int
    ) -> Tuple[Dict[int, Tuple[int, int]], int]:
        """Which (file -> row range) this rank reads for an even global split."""
        total = sum(lengths)
        per_rank = total // world_size
        remainder = total % world_size
        start = rank * per_rank
        end = start + per_rank
        out: Dict[int, Tuple[int, int]] = {}
        file_start = 0
        for i, n in enumerate(lengths):
            file_end = file_start + n
            lo = max(start, file_start)
            hi = min(end, file_end)
            if hi > lo:
                out[i] = (lo - file_start, hi - file_start)
            file_start = file_end
        return out, remainder


class InMemoryBinaryCriteoIterDataPipe(IterableDataset):
    """Batched iterator over preprocessed .npy shards, one slice per rank
    (reference criteo.py:715)."""

    def __init__(
        self,
        dense_paths: List[str],
        sparse_paths: List[str],
        labels_paths: List[str],
        batch_size: int,
        rank: int = 0,
        world_size: int = 1,
        hashes: Optional[List[int]] = None,
        shuffle_batches: bool = False,
        seed: int = 0,
    ) -> None:
        self.batch_size = batch_size
        self.rank = rank
        self.world_size = world_size
        self.hashes = hashes
        self.shuffle_batches = shuffle_batches
        self.seed = seed
        lengths = [BinaryCriteoUtils.get_shape_from_npy(p)[0] for p in dense_paths]
        ranges, _ = BinaryCriteoUtils.get_file_row_ranges_and_remainder(
            lengths, rank, world_size
        )
        dense_l, sparse_l, labels_l = [], [], []
        for i, (lo, hi) in ranges.items():
            dense_l.append(np.load(dense_paths[i], mmap_mode="r")[lo:hi])
            sparse_l.append(np.load(sparse_paths[i], mmap_mode="r")[lo:hi])
            labels_l.append(np.load(labels_paths[i], mmap_mode="r")[lo:hi])
        self.dense = np.concatenate(dense_l) if dense_l else np.zeros((0, INT_FEATURE_COUNT), np.float32)
        self.sparse = np.concatenate(sparse_l) if sparse_l else np.zeros((0, CAT_FEATURE_COUNT), np.int64)
        self.labels = np.concatenate(labels_l) if labels_l else np.zeros((0, 1), np.int32)
        if self.hashes is not None:
            self.sparse = self.sparse % np.array(self.hashes, dtype=np.int64)

    def __len__(self) -> int:
        return (self.dense.shape[0] + self.batch_size - 1) // self.batch_size

    def _make_batch(self, dense: np.ndarray, sparse: np.ndarray, labels: np.ndarray) -> Batch:
        B = dense.shape[0]
        values = torch.from_numpy(np.ascontiguousarray(sparse.T.reshape(-1)))
        kjt = KeyedJaggedTensor(
            keys=DEFAULT_CAT_NAMES,
            values=values,
            lengths=torch.ones(CAT_FEATURE_COUNT * B, dtype=torch.int64),
            stride=B,
        )
        return Batch(
            dense_features=torch.from_numpy(np.ascontiguousarray(dense)).float().log1p(),
            sparse_features=kjt,
            labels=torch.from_numpy(np.ascontiguousarray(labels.reshape(-1))).int(),
        )

    def __iter__(self) -> Iterator[Batch]:
        n = self.dense.shape[0]
        order = np.arange(n)
        if self.shuffle_batches:
            np.random.default_rng(self.seed).shuffle(order)
        for start in range(0, n, self.batch_size):
            idx = order[start : start + self.batch_size]
            yield self._make_batch(self.dense[idx], self.sparse[idx], self.labels[idx])
