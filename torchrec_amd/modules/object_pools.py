"""Object pools: batched storage keyed by stable ids.

Reference parity: torchrec/modules/tensor_pool.py:28 (TensorPool) and
torchrec/modules/keyed_jagged_tensor_pool.py:77 (KeyedJaggedTensorPool) —
used for cached sequence state in retrieval models.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

from torchrec_amd import ops
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


class TensorPool(nn.Module):
    """Fixed-capacity [pool_size, dim] store with id-indexed lookup/update."""

    def __init__(
        self,
        pool_size: int,
        dim: int,
        dtype: torch.dtype = torch.float32,
        device: Optional[torch.device] = None,
        enable_uvm: bool = False,
    ) -> None:
        super().__init__()
        device = device or torch.device("cpu")
        pool = torch.zeros(pool_size, dim, dtype=dtype, device="cpu" if enable_uvm else device)
        if enable_uvm and device.type == "cuda":
            pool = pool.pin_memory()
        self.register_buffer("_pool", pool)
        self._pool_size = pool_size
        self._dim = dim

    @property
    def pool_size(self) -> int:
        return self._pool_size

    @property
    def dim(self) -> int:
        return self._dim

    def lookup(self, ids: torch.Tensor) -> torch.Tensor:
        return self._pool[ids.to(self._pool.device)].to(ids.device)

    def update(self, ids: torch.Tensor, values: torch.Tensor) -> None:
        self._pool[ids.to(self._pool.device)] = values.to(self._pool.dtype).to(
            self._pool.device
        )

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        return self.lookup(ids)


class KeyedJaggedTensorPool(nn.Module):
    """Fixed-capacity store of per-id jagged feature lists.

    Each slot holds up to ``feature_max_lengths[k]`` values per feature; rows
    are stored padded-dense internally and converted back to KJT on lookup.
    """

    def __init__(
        self,
        pool_size: int,
        feature_max_lengths: dict,
        values_dtype: torch.dtype = torch.int64,
        device: Optional[torch.device] = None,
    ) -> None:
        super().__init__()
        device = device or torch.device("cpu")
        self._pool_size = pool_size
        self._feature_max_lengths = dict(feature_max_lengths)
        self._keys = list(feature_max_lengths.keys())
        total = sum(feature_max_lengths.values())
        self.register_buffer(
            "_values", torch.zeros(pool_size, total, dtype=values_dtype, device=device)
        )
        self.register_buffer(
            "_lengths",
            torch.zeros(pool_size, len(self._keys), dtype=torch.int64, device=device),
        )
        offs = [0]
        for k in self._keys:
            offs.append(offs[-1] + feature_max_lengths[k])
        self._col_offsets = offs

    @property
    def pool_size(self) -> int:
        return self._pool_size

    def update(self, ids: torch.Tensor, values: KeyedJaggedTensor) -> None:
        assert values.keys() == self._keys, "feature keys must match pool schema"
        B = values.stride()
        jts = values.to_dict()
        for ki, k in enumerate(self._keys):
            jt = jts[k]
            maxlen = self._feature_max_lengths[k]
            dense = ops.jagged_to_padded_dense(
                jt.values().unsqueeze(1).float(), jt.offsets(), maxlen, 0.0
            ).squeeze(-1)
            self._values[ids, self._col_offsets[ki] : self._col_offsets[ki] + maxlen] = (
                dense.to(self._values.dtype)
            )
            self._lengths[ids, ki] = jt.lengths().clamp(max=maxlen)

    def lookup(self, ids: torch.Tensor) -> KeyedJaggedTensor:
        B = ids.numel()
        values_list: List[torch.Tensor] = []
        lengths_list: List[torch.Tensor] = []
        for ki, k in enumerate(self._keys):
            maxlen = self._feature_max_lengths[k]
            dense = self._values[ids, self._col_offsets[ki] : self._col_offsets[ki] + maxlen]
            lengths = self._lengths[ids, ki]
            mask = torch.arange(maxlen, device=dense.device).expand(B, -1) < lengths.unsqueeze(1)
            values_list.append(dense[mask])
            lengths_list.append(lengths)
        return KeyedJaggedTensor(
            keys=self._keys,
            values=torch.cat(values_list) if values_list else self._values.new_empty(0),
            lengths=torch.cat(lengths_list),
            stride=B,
        )

    def forward(self, ids: torch.Tensor) -> KeyedJaggedTensor:
        return self.lookup(ids)
