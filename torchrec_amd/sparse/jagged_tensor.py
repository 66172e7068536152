"""Jagged sparse tensor types — the framework's core data structures.

API surface mirrors the reference's ``torchrec/sparse/jagged_tensor.py``
(JaggedTensor :638, KeyedJaggedTensor :1913, KeyedTensor :3518) so reference
users can switch; the implementation is fresh and routes all hot paths through
``torchrec_amd.ops`` (CPU reference impls / CDNA4 HIP kernels).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from torchrec_amd import ops
from torchrec_amd.streamable import Pipelineable


def _maybe_compute_offsets(
    lengths: Optional[torch.Tensor], offsets: Optional[torch.Tensor]
) -> torch.Tensor:
    if offsets is not None:
        return offsets
    assert lengths is not None, "need lengths or offsets"
    return ops.complete_cumsum(lengths)


def _maybe_compute_lengths(
    lengths: Optional[torch.Tensor], offsets: Optional[torch.Tensor]
) -> torch.Tensor:
    if lengths is not None:
        return lengths
    assert offsets is not None, "need lengths or offsets"
    return offsets[1:] - offsets[:-1]


class JaggedTensor(Pipelineable):
    """values + lengths/offsets (+ optional per-value weights).

    Reference parity: torchrec/sparse/jagged_tensor.py:638.
    """

    def __init__(
        self,
        values: torch.Tensor,
        weights: Optional[torch.Tensor] = None,
        lengths: Optional[torch.Tensor] = None,
        offsets: Optional[torch.Tensor] = None,
    ) -> None:
        self._values = values
        self._weights = weights
        assert lengths is not None or offsets is not None, (
            "JaggedTensor requires lengths or offsets"
        )
        self._lengths = lengths
        self._offsets = offsets

    @staticmethod
    def empty(
        is_weighted: bool = False,
        device: Optional[torch.device] = None,
        values_dtype: Optional[torch.dtype] = None,
        weights_dtype: Optional[torch.dtype] = None,
        lengths_dtype: torch.dtype = torch.int64,
    ) -> "JaggedTensor":
        return JaggedTensor(
            values=torch.empty(0, dtype=values_dtype, device=device),
            weights=(
                torch.empty(0, dtype=weights_dtype, device=device) if is_weighted else None
            ),
            lengths=torch.empty(0, dtype=lengths_dtype, device=device),
        )

    @staticmethod
    def from_dense_lengths(
        values: torch.Tensor,
        lengths: torch.Tensor,
        weights: Optional[torch.Tensor] = None,
    ) -> "JaggedTensor":
        """[B, N] dense + per-row lengths -> jagged."""
        mask = torch.arange(values.shape[1], device=values.device).expand(
            values.shape[0], -1
        ) < lengths.unsqueeze(1)
        return JaggedTensor(
            values=values[mask],
            weights=weights[mask] if weights is not None else None,
            lengths=lengths.to(torch.int64),
        )

    @staticmethod
    def from_dense(
        values: List[torch.Tensor],
        weights: Optional[List[torch.Tensor]] = None,
    ) -> "JaggedTensor":
        lengths = torch.tensor([v.numel() for v in values], dtype=torch.int64)
        return JaggedTensor(
            values=torch.cat(values) if values else torch.empty(0),
            weights=torch.cat(weights) if weights is not None else None,
            lengths=lengths,
        )

    def lengths(self) -> torch.Tensor:
        self._lengths = _maybe_compute_lengths(self._lengths, self._offsets)
        return self._lengths

    def lengths_or_none(self) -> Optional[torch.Tensor]:
        return self._lengths

    def offsets(self) -> torch.Tensor:
        self._offsets = _maybe_compute_offsets(self._lengths, self._offsets)
        return self._offsets

    def offsets_or_none(self) -> Optional[torch.Tensor]:
        return self._offsets

    def values(self) -> torch.Tensor:
        return self._values

    def weights(self) -> torch.Tensor:
        assert self._weights is not None, "JaggedTensor has no weights"
        return self._weights

    def weights_or_none(self) -> Optional[torch.Tensor]:
        return self._weights

    def device(self) -> torch.device:
        return self._values.device

    def to_dense(self) -> List[torch.Tensor]:
        offsets = self.offsets().tolist()
        return [self._values[offsets[i] : offsets[i + 1]] for i in range(len(offsets) - 1)]

    def to_dense_weights(self) -> Optional[List[torch.Tensor]]:
        if self._weights is None:
            return None
        offsets = self.offsets().tolist()
        return [self._weights[offsets[i] : offsets[i + 1]] for i in range(len(offsets) - 1)]

    def to_padded_dense(
        self, desired_length: Optional[int] = None, padding_value: float = 0.0
    ) -> torch.Tensor:
        """Reference parity: jagged_tensor.py:956 (fbgemm.jagged_to_padded_dense)."""
        N = desired_length if desired_length is not None else int(self.lengths().max().item())
        return ops.jagged_to_padded_dense(self._values, self.offsets(), N, padding_value)

    def to(self, device: torch.device, non_blocking: bool = False) -> "JaggedTensor":
        return JaggedTensor(
            values=self._values.to(device, non_blocking=non_blocking),
            weights=(
                self._weights.to(device, non_blocking=non_blocking)
                if self._weights is not None
                else None
            ),
            lengths=(
                self._lengths.to(device, non_blocking=non_blocking)
                if self._lengths is not None
                else None
            ),
            offsets=(
                self._offsets.to(device, non_blocking=non_blocking)
                if self._offsets is not None
                else None
            ),
        )

    def record_stream(self, stream: torch.Stream) -> None:
        self._values.record_stream(stream)
        for t in (self._weights, self._lengths, self._offsets):
            if t is not None:
                t.record_stream(stream)

    def __str__(self) -> str:
        return f"JaggedTensor(values={self._values}, lengths={self._lengths})"


def _kjt_empty_like(device: torch.device, values_dtype: torch.dtype) -> "KeyedJaggedTensor":
    return KeyedJaggedTensor(
        keys=[],
        values=torch.empty(0, dtype=values_dtype, device=device),
        lengths=torch.empty(0, dtype=torch.int64, device=device),
        stride=0,
    )


class KeyedJaggedTensor(Pipelineable):
    """A batch of jagged id-lists keyed by feature name — THE input type.

    Storage is feature-major: ``values`` holds feature k0's ids for samples
    0..B-1, then k1's, ...; ``lengths`` is [K * B] feature-major; ``stride``
    is the batch size B.

    Reference parity: torchrec/sparse/jagged_tensor.py:1913 (split :2676,
    permute :2831, to_dict :3038, a2a plumbing :3207-3257).
    """

    def __init__(
        self,
        keys: List[str],
        values: torch.Tensor,
        weights: Optional[torch.Tensor] = None,
        lengths: Optional[torch.Tensor] = None,
        offsets: Optional[torch.Tensor] = None,
        stride: Optional[int] = None,
        stride_per_key_per_rank: Optional[List[List[int]]] = None,
        length_per_key: Optional[List[int]] = None,
        offset_per_key: Optional[List[int]] = None,
        index_per_key: Optional[Dict[str, int]] = None,
        jt_dict: Optional[Dict[str, JaggedTensor]] = None,
    ) -> None:
        self._keys = keys
        self._values = values
        self._weights = weights
        self._lengths = lengths
        self._offsets = offsets
        self._stride_per_key_per_rank = stride_per_key_per_rank
        self._variable_stride_per_key = stride_per_key_per_rank is not None
        if stride is not None:
            self._stride = stride
        elif stride_per_key_per_rank is not None:
            self._stride = (
                max(sum(s) for s in stride_per_key_per_rank) if stride_per_key_per_rank else 0
            )
        else:
            n = lengths.numel() if lengths is not None else (offsets.numel() - 1)
            self._stride = n // len(keys) if len(keys) else 0
        self._length_per_key = length_per_key
        self._offset_per_key = offset_per_key
        self._index_per_key = index_per_key
        self._jt_dict = jt_dict

    # -- constructors ------------------------------------------------------

    @staticmethod
    def from_lengths_sync(
        keys: List[str],
        values: torch.Tensor,
        lengths: torch.Tensor,
        weights: Optional[torch.Tensor] = None,
        stride: Optional[int] = None,
    ) -> "KeyedJaggedTensor":
        kjt = KeyedJaggedTensor(
            keys=keys, values=values, weights=weights, lengths=lengths, stride=stride
        )
        return kjt.sync()

    @staticmethod
    def from_offsets_sync(
        keys: List[str],
        values: torch.Tensor,
        offsets: torch.Tensor,
        weights: Optional[torch.Tensor] = None,
        stride: Optional[int] = None,
    ) -> "KeyedJaggedTensor":
        kjt = KeyedJaggedTensor(
            keys=keys, values=values, weights=weights, offsets=offsets, stride=stride
        )
        return kjt.sync()

    @staticmethod
    def from_jt_dict(jt_dict: Dict[str, JaggedTensor]) -> "KeyedJaggedTensor":
        keys = list(jt_dict.keys())
        values = torch.cat([jt_dict[k].values() for k in keys])
        lengths = torch.cat([jt_dict[k].lengths() for k in keys])
        has_w = any(jt_dict[k].weights_or_none() is not None for k in keys)
        weights = torch.cat([jt_dict[k].weights() for k in keys]) if has_w else None
        return KeyedJaggedTensor(keys=keys, values=values, weights=weights, lengths=lengths)

    @staticmethod
    def empty(
        is_weighted: bool = False,
        device: Optional[torch.device] = None,
        values_dtype: Optional[torch.dtype] = None,
        weights_dtype: Optional[torch.dtype] = None,
        lengths_dtype: torch.dtype = torch.int64,
        stride: int = 0,
    ) -> "KeyedJaggedTensor":
        return KeyedJaggedTensor(
            keys=[],
            values=torch.empty(0, dtype=values_dtype, device=device),
            weights=torch.empty(0, dtype=weights_dtype, device=device) if is_weighted else None,
            lengths=torch.empty(0, dtype=lengths_dtype, device=device),
            stride=stride,
        )

    @staticmethod
    def concat(kjt_list: List["KeyedJaggedTensor"]) -> "KeyedJaggedTensor":
        """Concatenate along the key dimension (same stride)."""
        keys: List[str] = []
        values, lengths, weights = [], [], []
        has_w = any(k._weights is not None for k in kjt_list)
        stride = kjt_list[0].stride()
        for k in kjt_list:
            assert k.stride() == stride
            keys.extend(k.keys())
            values.append(k.values())
            lengths.append(k.lengths())
            if has_w:
                weights.append(k.weights())
        return KeyedJaggedTensor(
            keys=keys,
            values=torch.cat(values),
            lengths=torch.cat(lengths),
            weights=torch.cat(weights) if has_w else None,
            stride=stride,
        )

    # -- accessors ---------------------------------------------------------

    def keys(self) -> List[str]:
        return self._keys

    def values(self) -> torch.Tensor:
        return self._values

    def weights(self) -> torch.Tensor:
        assert self._weights is not None, "KJT has no weights"
        return self._weights

    def weights_or_none(self) -> Optional[torch.Tensor]:
        return self._weights

    def lengths(self) -> torch.Tensor:
        self._lengths = _maybe_compute_lengths(self._lengths, self._offsets)
        return self._lengths

    def lengths_or_none(self) -> Optional[torch.Tensor]:
        return self._lengths

    def offsets(self) -> torch.Tensor:
        self._offsets = _maybe_compute_offsets(self._lengths, self._offsets)
        return self._offsets

    def offsets_or_none(self) -> Optional[torch.Tensor]:
        return self._offsets

    def stride(self) -> int:
        return self._stride

    def variable_stride_per_key(self) -> bool:
        return self._variable_stride_per_key

    def stride_per_key_per_rank(self) -> List[List[int]]:
        return self._stride_per_key_per_rank or [[self._stride]] * len(self._keys)

    def stride_per_key(self) -> List[int]:
        if self._variable_stride_per_key:
            return [sum(s) for s in self._stride_per_key_per_rank]
        return [self._stride] * len(self._keys)

    def device(self) -> torch.device:
        return self._values.device

    def sync(self) -> "KeyedJaggedTensor":
        """Compute host-side length_per_key / offset_per_key (may D2H sync)."""
        self.length_per_key()
        self.offset_per_key()
        return self

    def unsync(self) -> "KeyedJaggedTensor":
        self._length_per_key = None
        self._offset_per_key = None
        return self

    def length_per_key(self) -> List[int]:
        if self._length_per_key is None:
            if len(self._keys) == 0:
                self._length_per_key = []
            elif self._variable_stride_per_key:
                # per-key lengths segments have per-key sizes (VBE)
                key_strides = [sum(sp) for sp in self.stride_per_key_per_rank()]
                self._length_per_key = [
                    int(seg.sum())
                    for seg in self.lengths().split(key_strides)
                ]
            else:
                lpk = self.lengths().view(len(self._keys), -1).sum(dim=1)
                self._length_per_key = lpk.cpu().tolist()
        return self._length_per_key

    def length_per_key_or_none(self) -> Optional[List[int]]:
        return self._length_per_key

    def offset_per_key(self) -> List[int]:
        if self._offset_per_key is None:
            opk = [0]
            for l in self.length_per_key():
                opk.append(opk[-1] + l)
            self._offset_per_key = opk
        return self._offset_per_key

    def offset_per_key_or_none(self) -> Optional[List[int]]:
        return self._offset_per_key

    def _key_indices(self) -> Dict[str, int]:
        if self._index_per_key is None:
            self._index_per_key = {k: i for i, k in enumerate(self._keys)}
        return self._index_per_key

    # -- transforms --------------------------------------------------------

    def split(self, segments: List[int]) -> List["KeyedJaggedTensor"]:
        """Split by consecutive key groups (zero-copy views on values).

        Reference parity: jagged_tensor.py:2676.
        """
        assert sum(segments) == len(self._keys)
        out: List[KeyedJaggedTensor] = []
        start = 0
        opk = self.offset_per_key()
        if self._variable_stride_per_key:
            # per-key lengths segments have per-key sizes (VBE)
            spk = self.stride_per_key_per_rank()
            key_strides = [sum(s) for s in spk]
            len_bounds = [0]
            for ks in key_strides:
                len_bounds.append(len_bounds[-1] + ks)
            lengths = self.lengths()
            for seg in segments:
                keys = self._keys[start : start + seg]
                vstart, vend = opk[start], opk[start + seg]
                out.append(
                    KeyedJaggedTensor(
                        keys=keys,
                        values=self._values[vstart:vend],
                        weights=self._weights[vstart:vend]
                        if self._weights is not None
                        else None,
                        lengths=lengths[len_bounds[start] : len_bounds[start + seg]],
                        stride_per_key_per_rank=spk[start : start + seg],
                        length_per_key=self._length_per_key[start : start + seg]
                        if self._length_per_key is not None
                        else None,
                    )
                )
                start += seg
            return out
        B = self._stride
        lengths = self.lengths()
        for seg in segments:
            keys = self._keys[start : start + seg]
            vstart, vend = opk[start], opk[start + seg]
            out.append(
                KeyedJaggedTensor(
                    keys=keys,
                    values=self._values[vstart:vend],
                    weights=self._weights[vstart:vend] if self._weights is not None else None,
                    lengths=lengths[start * B : (start + seg) * B],
                    stride=B,
                    length_per_key=self._length_per_key[start : start + seg]
                    if self._length_per_key is not None
                    else None,
                )
            )
            start += seg
        return out

    def permute(
        self, indices: List[int], indices_tensor: Optional[torch.Tensor] = None
    ) -> "KeyedJaggedTensor":
        """Reorder (and optionally duplicate) features.

        Reference parity: jagged_tensor.py:2831 (fbgemm permute_2D_sparse_data
        at :2913).
        """
        if indices_tensor is None:
            indices_tensor = torch.tensor(indices, dtype=torch.int64, device=self.device())
        if self._variable_stride_per_key:
            # VBE: per-key jagged segments; value counts stay on device
            spk = self.stride_per_key_per_rank()
            key_strides = [sum(sp) for sp in spk]
            lengths = self.lengths()
            loffs = torch.zeros(
                lengths.numel() + 1, dtype=torch.int64, device=lengths.device
            )
            torch.cumsum(lengths, 0, out=loffs[1:])
            bounds = [0]
            for ks in key_strides:
                bounds.append(bounds[-1] + ks)
            bt = torch.tensor(bounds, device=lengths.device)
            seg_vals = loffs[bt[1:]] - loffs[bt[:-1]]
            _pl, pv, pw = ops.permute_2d_sparse_data(
                indices_tensor,
                seg_vals.view(-1, 1),
                self._values,
                self._weights,
                permuted_lengths_sum=int(self._values.numel()),
            )
            len_pieces = list(lengths.split(key_strides))
            new_lengths = torch.cat([len_pieces[i] for i in indices])
            return KeyedJaggedTensor(
                keys=[self._keys[i] for i in indices],
                values=pv,
                weights=pw,
                lengths=new_lengths,
                stride_per_key_per_rank=[spk[i] for i in indices],
            )
        B = self._stride
        lengths2d = self.lengths().view(len(self._keys), B)
        pl, pv, pw = ops.permute_2d_sparse_data(
            indices_tensor, lengths2d, self._values, self._weights
        )
        lpk = (
            [self._length_per_key[i] for i in indices]
            if self._length_per_key is not None
            else None
        )
        return KeyedJaggedTensor(
            keys=[self._keys[i] for i in indices],
            values=pv,
            weights=pw,
            lengths=pl.reshape(-1),
            stride=B,
            length_per_key=lpk,
        )

    def __getitem__(self, key: str) -> JaggedTensor:
        i = self._key_indices()[key]
        opk = self.offset_per_key()
        if self._variable_stride_per_key:
            key_strides = [sum(sp) for sp in self.stride_per_key_per_rank()]
            lo = sum(key_strides[:i])
            lengths_i = self.lengths()[lo : lo + key_strides[i]]
        else:
            B = self._stride
            lengths_i = self.lengths()[i * B : (i + 1) * B]
        return JaggedTensor(
            values=self._values[opk[i] : opk[i + 1]],
            weights=self._weights[opk[i] : opk[i + 1]] if self._weights is not None else None,
            lengths=lengths_i,
        )

    def to_dict(self) -> Dict[str, JaggedTensor]:
        """Reference parity: jagged_tensor.py:3038."""
        if self._jt_dict is None:
            self._jt_dict = {k: self[k] for k in self._keys}
        return self._jt_dict

    def flatten_lengths(self) -> "KeyedJaggedTensor":
        return self

    def to(self, device: torch.device, non_blocking: bool = False) -> "KeyedJaggedTensor":
        return KeyedJaggedTensor(
            keys=self._keys,
            values=self._values.to(device, non_blocking=non_blocking),
            weights=(
                self._weights.to(device, non_blocking=non_blocking)
                if self._weights is not None
                else None
            ),
            lengths=(
                self._lengths.to(device, non_blocking=non_blocking)
                if self._lengths is not None
                else None
            ),
            offsets=(
                self._offsets.to(device, non_blocking=non_blocking)
                if self._offsets is not None
                else None
            ),
            stride=self._stride,
            stride_per_key_per_rank=self._stride_per_key_per_rank,
            length_per_key=self._length_per_key,
            offset_per_key=self._offset_per_key,
            index_per_key=self._index_per_key,
        )

    def pin_memory(self) -> "KeyedJaggedTensor":
        return KeyedJaggedTensor(
            keys=self._keys,
            values=self._values.pin_memory(),
            weights=self._weights.pin_memory() if self._weights is not None else None,
            lengths=self._lengths.pin_memory() if self._lengths is not None else None,
            offsets=self._offsets.pin_memory() if self._offsets is not None else None,
            stride=self._stride,
            stride_per_key_per_rank=self._stride_per_key_per_rank,
            length_per_key=self._length_per_key,
            offset_per_key=self._offset_per_key,
        )

    def record_stream(self, stream: torch.Stream) -> None:
        self._values.record_stream(stream)
        for t in (self._weights, self._lengths, self._offsets):
            if t is not None:
                t.record_stream(stream)

    def __str__(self) -> str:
        return (
            f"KeyedJaggedTensor(keys={self._keys}, stride={self._stride}, "
            f"values.shape={tuple(self._values.shape)})"
        )


class KeyedTensor(Pipelineable):
    """Dense [B, sum_D] concat of pooled embeddings keyed by name.

    Reference parity: torchrec/sparse/jagged_tensor.py:3518 (regroup :268-318).
    """

    def __init__(
        self,
        keys: List[str],
        length_per_key: List[int],
        values: torch.Tensor,
        key_dim: int = 1,
        offset_per_key: Optional[List[int]] = None,
        index_per_key: Optional[Dict[str, int]] = None,
    ) -> None:
        self._keys = keys
        self._length_per_key = length_per_key
        self._values = values
        self._key_dim = key_dim
        self._offset_per_key = offset_per_key
        self._index_per_key = index_per_key

    @staticmethod
    def from_tensor_list(
        keys: List[str], tensors: List[torch.Tensor], key_dim: int = 1, cat_dim: int = 1
    ) -> "KeyedTensor":
        length_per_key = [t.shape[key_dim] for t in tensors]
        return KeyedTensor(
            keys=keys,
            length_per_key=length_per_key,
            values=torch.cat(tensors, dim=cat_dim),
            key_dim=key_dim,
        )

    def keys(self) -> List[str]:
        return self._keys

    def values(self) -> torch.Tensor:
        return self._values

    def key_dim(self) -> int:
        return self._key_dim

    def device(self) -> torch.device:
        return self._values.device

    def length_per_key(self) -> List[int]:
        return self._length_per_key

    def offset_per_key(self) -> List[int]:
        if self._offset_per_key is None:
            opk = [0]
            for l in self._length_per_key:
                opk.append(opk[-1] + l)
            self._offset_per_key = opk
        return self._offset_per_key

    def _key_indices(self) -> Dict[str, int]:
        if self._index_per_key is None:
            self._index_per_key = {k: i for i, k in enumerate(self._keys)}
        return self._index_per_key

    def __getitem__(self, key: str) -> torch.Tensor:
        i = self._key_indices()[key]
        opk = self.offset_per_key()
        return self._values.narrow(self._key_dim, opk[i], self._length_per_key[i])

    def to_dict(self) -> Dict[str, torch.Tensor]:
        return {k: self[k] for k in self._keys}

    @staticmethod
    def regroup(
        keyed_tensors: List["KeyedTensor"], groups: List[List[str]]
    ) -> List[torch.Tensor]:
        """Regroup columns of several KTs into new groups (differentiable).

        Reference parity: jagged_tensor.py:268-318 (permute_multi_embedding).
        """
        key_to_src: Dict[str, Tuple[int, int]] = {}
        for ti, kt in enumerate(keyed_tensors):
            for ki, k in enumerate(kt.keys()):
                key_to_src[k] = (ti, ki)
        out = []
        for group in groups:
            cols = []
            for k in group:
                ti, _ = key_to_src[k]
                cols.append(keyed_tensors[ti][k])
            out.append(torch.cat(cols, dim=1))
        return out

    @staticmethod
    def regroup_as_dict(
        keyed_tensors: List["KeyedTensor"], groups: List[List[str]], keys: List[str]
    ) -> Dict[str, torch.Tensor]:
        tensors = KeyedTensor.regroup(keyed_tensors, groups)
        return dict(zip(keys, tensors))

    def to(self, device: torch.device, non_blocking: bool = False) -> "KeyedTensor":
        return KeyedTensor(
            keys=self._keys,
            length_per_key=self._length_per_key,
            values=self._values.to(device, non_blocking=non_blocking),
            key_dim=self._key_dim,
            offset_per_key=self._offset_per_key,
        )

    def record_stream(self, stream: torch.Stream) -> None:
        self._values.record_stream(stream)

    def __str__(self) -> str:
        return f"KeyedTensor(keys={self._keys}, values.shape={tuple(self._values.shape)})"
