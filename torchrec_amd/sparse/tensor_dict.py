"""TensorDict interop (reference: torchrec/sparse/tensor_dict.py
maybe_td_to_kjt). tensordict is optional — absent, inputs pass through."""

from __future__ import annotations

from typing import Any, List, Optional

import torch

from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


def maybe_td_to_kjt(features: Any, keys: Optional[List[str]] = None):
    """If ``features`` is a tensordict.TensorDict of (values, lengths) nested
    tensors, convert to a KJT; any other input passes through unchanged."""
    try:
        from tensordict import TensorDict  # type: ignore
    except ImportError:
        return features
    if not isinstance(features, TensorDict):
        return features
    keys = keys or list(features.keys())
    values, lengths = [], []
    for k in keys:
        jt = features[k]
        values.append(jt["values"])
        lengths.append(jt["lengths"])
    return KeyedJaggedTensor(
        keys=keys, values=torch.cat(values), lengths=torch.cat(lengths)
    )
