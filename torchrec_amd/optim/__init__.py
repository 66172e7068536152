"""Optimizer surface (reference: torchrec/optim/__init__.py)."""

from torchrec_amd.optim.keyed import (  # noqa: F401
    CombinedOptimizer,
    FusedOptimizer,
    KeyedOptimizer,
)
