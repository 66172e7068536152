"""TensorDict-style interop (reference: torchrec/sparse/tensor_dict.py
maybe_td_to_kjt): accept a plain dict of per-feature tensors/JTs as model
input and convert to KJT."""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import torch

from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor


def maybe_td_to_kjt(features: Any, keys: Optional[List[str]] = None) -> KeyedJaggedTensor:
    if isinstance(features, KeyedJaggedTensor):
        return features
    if isinstance(features, dict):
        jts: Dict[str, JaggedTensor] = {}
        for k in keys or features.keys():
            v = features[k]
            if isinstance(v, JaggedTensor):
                jts[k] = v
            elif isinstance(v, torch.Tensor) and v.dim() == 2:
                # dense [B, L] of ids -> fixed-length jagged
                B, L = v.shape
                jts[k] = JaggedTensor(
                    values=v.reshape(-1),
                    lengths=torch.full((B,), L, dtype=torch.int64, device=v.device),
                )
            elif isinstance(v, torch.Tensor) and v.dim() == 1:
                jts[k] = JaggedTensor(
                    values=v,
                    lengths=torch.ones(v.numel(), dtype=torch.int64, device=v.device),
                )
            else:
                raise TypeError(f"cannot convert feature {k!r} of type {type(v)}")
        return KeyedJaggedTensor.from_jt_dict(jts)
    raise TypeError(f"cannot convert {type(features)} to KeyedJaggedTensor")
