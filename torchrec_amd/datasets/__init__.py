"""Datasets (reference: torchrec/datasets/__init__.py)."""

from torchrec_amd.datasets.random import Batch, RandomRecDataset, generate_batch  # noqa: F401
