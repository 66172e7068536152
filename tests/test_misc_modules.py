"""Coverage for the small leaf modules: streamable contracts, the MovieLens
loader, and module-level toggles."""

import csv
import tempfile

import torch

from torchrec_amd.datasets.movielens import movielens_20m, movielens_25m
from torchrec_amd.datasets.random import Batch
from torchrec_amd.distributed.global_settings import (
    get_propagate_device,
    set_propagate_device,
)
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor
from torchrec_amd.streamable import Multistreamable, Pipelineable


def test_movielens_reader():
    with tempfile.TemporaryDirectory() as d:
        with open(f"{d}/ratings.csv", "w", newline="") as f:
            w = csv.writer(f)
            w.writerow(["userId", "movieId", "rating", "timestamp"])
            w.writerow([1, 31, 2.5, 1112486027])
            w.writerow([2, 1029, 3.0, 1112484676])
        rows = list(movielens_20m(d))
        assert rows == [
            {"userId": 1, "movieId": 31, "rating": 2.5, "timestamp": 1112486027},
            {"userId": 2, "movieId": 1029, "rating": 3.0, "timestamp": 1112484676},
        ]
        mapped = list(movielens_25m(d, row_mapper=lambda r: r["movieId"]))
        assert mapped == [31, 1029]


def test_global_settings_toggle():
    assert get_propagate_device() is False
    set_propagate_device(True)
    try:
        assert get_propagate_device() is True
    finally:
        set_propagate_device(False)


def test_batch_is_pipelineable():
    """The bench/dataset Batch satisfies the Multistreamable/Pipelineable
    contract the pipelines rely on (to + record_stream forwarding)."""
    kjt = KeyedJaggedTensor(
        keys=["f0"],
        values=torch.tensor([1, 2, 3]),
        lengths=torch.tensor([1, 2]),
        stride=2,
    )
    b = Batch(torch.randn(2, 4), kjt, torch.tensor([0.0, 1.0]))
    assert isinstance(b, Pipelineable) and isinstance(b, Multistreamable)
    b2 = b.to(torch.device("cpu"), non_blocking=False)
    assert torch.equal(b2.dense_features, b.dense_features)
    # record_stream on CPU tensors is a no-op but must not raise on the
    # contract surface (GPU pipelines call it on every batch)
    assert callable(getattr(b, "record_stream"))
    if torch.cuda.is_available():  # pragma: no cover — GPU-only contract call
        b.record_stream(torch.cuda.current_stream())
