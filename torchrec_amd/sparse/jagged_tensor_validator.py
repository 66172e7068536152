"""KJT invariant validation (reference: torchrec/sparse/jagged_tensor_validator.py)."""

from __future__ import annotations

import torch

from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor


def validate_jagged_tensor(jt: JaggedTensor) -> None:
    lengths = jt.lengths()
    assert (lengths >= 0).all(), "negative lengths"
    assert int(lengths.sum()) == jt.values().numel() or jt.values().dim() == 2, (
        "values/lengths mismatch"
    )
    if jt.weights_or_none() is not None:
        assert jt.weights().numel() == jt.values().shape[0] if jt.values().dim() == 2 else jt.values().numel()
    offsets = jt.offsets()
    assert (offsets[1:] >= offsets[:-1]).all(), "offsets must be non-decreasing"


def validate_keyed_jagged_tensor(kjt: KeyedJaggedTensor) -> None:
    K = len(kjt.keys())
    assert len(set(kjt.keys())) == K, "duplicate keys"
    lengths = kjt.lengths()
    if not kjt.variable_stride_per_key():
        assert lengths.numel() == K * kjt.stride(), (
            f"lengths numel {lengths.numel()} != keys {K} * stride {kjt.stride()}"
        )
    assert (lengths >= 0).all(), "negative lengths"
    assert int(lengths.sum()) == kjt.values().numel(), "values/lengths mismatch"
    if kjt.weights_or_none() is not None:
        assert kjt.weights().numel() == kjt.values().numel(), "weights/values mismatch"
