"""MovieLens dataset loader (reference: torchrec/datasets/movielens.py)."""

from __future__ import annotations

import csv
from typing import Any, Dict, Iterator, List, Optional

import torch
from torch.utils.data import IterableDataset

DEFAULT_RATINGS_COLUMN_NAMES: List[str] = ["userId", "movieId", "rating", "timestamp"]


class MovieLensIterDataPipe(IterableDataset):
    """ratings.csv reader yielding row dicts (ml-20m/ml-25m layout)."""

    def __init__(self, ratings_path: str, row_mapper=None) -> None:
        self._path = ratings_path
        self._row_mapper = row_mapper

    def __iter__(self) -> Iterator[Any]:
        with open(self._path, newline="") as f:
            reader = csv.reader(f)
            header = next(reader)
            for row in reader:
                rec: Dict[str, Any] = {
                    "userId": int(row[0]),
                    "movieId": int(row[1]),
                    "rating": float(row[2]),
                    "timestamp": int(row[3]),
                }
                yield self._row_mapper(rec) if self._row_mapper else rec


def movielens_20m(root: str, row_mapper=None) -> MovieLensIterDataPipe:
    return MovieLensIterDataPipe(f"{root}/ratings.csv", row_mapper)


def movielens_25m(root: str, row_mapper=None) -> MovieLensIterDataPipe:
    return MovieLensIterDataPipe(f"{root}/ratings.csv", row_mapper)
