"""Quantized (int8 rowwise) inference embedding modules.

Reference parity: torchrec/quant/embedding_modules.py
(quant EmbeddingBagCollection :346 with from_float / quantize_state_dict
:217; EmbeddingCollection :748) — backed by the CDNA4 int8 TBE kernels
(ops/csrc/quant_tbe.hip) instead of FBGEMM IntNBit.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from torchrec_amd import ops
from torchrec_amd.modules.embedding_configs import (
    DataType,
    EmbeddingBagConfig,
    EmbeddingConfig,
    PoolingType,
)
from torchrec_amd.modules.embedding_modules import (
    EmbeddingBagCollection as FloatEBC,
    EmbeddingCollection as FloatEC,
)
from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor, KeyedTensor


def int8_row_stride(D: int) -> int:
    return ((D + 4 + 15) // 16) * 16


def quantize_rowwise_int8(weights: torch.Tensor) -> torch.Tensor:
    """fp32 [R, D] -> packed int8 rows [R, stride] (CPU ref / HIP kernel)."""
    if weights.is_cuda:
        ops.hip_ops()
        return torch.ops.trec_amd.quantize_rowwise_int8(weights)
    R, D = weights.shape
    stride = int8_row_stride(D)
    out = torch.zeros(R, stride, dtype=torch.uint8)
    mn = weights.min(dim=1).values
    mx = weights.max(dim=1).values
    scale = (mx - mn) / 255.0
    inv = torch.where(scale > 0, 1.0 / scale, torch.zeros_like(scale))
    q = ((weights - mn.unsqueeze(1)) * inv.unsqueeze(1) + 0.5).clamp(0, 255).floor()
    out[:, :D] = q.to(torch.uint8)
    sb = torch.stack([scale.half(), mn.half()], dim=1)  # [R, 2] fp16
    out[:, D : D + 4] = sb.view(torch.uint8).reshape(R, 4)
    return out


def dequantize_rowwise_int8(packed: torch.Tensor, D: int) -> torch.Tensor:
    """Packed rows -> fp32 [R, D] using the STORED fp16 scale/bias."""
    R = packed.shape[0]
    q = packed[:, :D].float()
    sb = packed[:, D : D + 4].reshape(R, 2, 2).view(torch.float16).reshape(R, 2)
    scale = sb[:, 0].float()
    bias = sb[:, 1].float()
    return q * scale.unsqueeze(1) + bias.unsqueeze(1)


class QuantTableBatchedEmbeddingBags(nn.Module):
    """Inference-only int8 TBE (pooled). Flat packed buffer, per-table byte
    offsets; GPU path = tbe_forward_pooled_int8, CPU path = dequant reference."""

    def __init__(
        self,
        specs: List[Tuple[str, int, int]],  # (name, rows, dim)
        feature_table_map: Optional[List[int]] = None,
        pooling: PoolingType = PoolingType.SUM,
        device: Optional[torch.device] = None,
        location: str = "device",  # "device" (HBM) | "managed" (pinned host, QUANT_UVM)
    ) -> None:
        super().__init__()
        device = device or torch.device("cpu")
        self._uvm = location == "managed" and device.type == "cuda"
        qdevice = torch.device("cpu") if self._uvm else device
        self._specs = specs
        T = len(specs)
        self._feature_table_map = feature_table_map or list(range(T))
        F = len(self._feature_table_map)
        self._num_features = F
        self.pooling = pooling
        byte_offsets = [0]
        for (_, rows, dim) in specs:
            byte_offsets.append(byte_offsets[-1] + rows * int8_row_stride(dim))
        dims = [s[2] for s in specs]
        feat_dims = [dims[t] for t in self._feature_table_map]
        d_out = [0]
        for d in feat_dims:
            d_out.append(d_out[-1] + d)
        self._total_D = d_out[-1]
        self._max_D = max(dims) if dims else 0
        qw = torch.zeros(byte_offsets[-1], dtype=torch.uint8, device=qdevice)
        if getattr(self, "_uvm", False):
            qw = qw.pin_memory()
        self.register_buffer("qweights", qw)
        reg = lambda n, t: self.register_buffer(n, t.to(device), persistent=False)
        reg("_table_byte_offsets", torch.tensor(byte_offsets[:-1], dtype=torch.int64))
        reg("_dims_t", torch.tensor(dims, dtype=torch.int32))
        reg("_feat_table_t", torch.tensor(self._feature_table_map, dtype=torch.int32))
        reg("_d_out_offsets", torch.tensor(d_out, dtype=torch.int64))
        reg("_empty_f", torch.empty(0, dtype=torch.float32))

    def load_float_table(self, i: int, weights: torch.Tensor) -> None:
        name, rows, dim = self._specs[i]
        packed = quantize_rowwise_int8(weights.to(self.qweights.device).float())
        start = int(self._table_byte_offsets[i])
        self.qweights[start : start + packed.numel()].copy_(packed.reshape(-1))

    def packed_table(self, i: int) -> torch.Tensor:
        name, rows, dim = self._specs[i]
        start = int(self._table_byte_offsets[i])
        return self.qweights[start : start + rows * int8_row_stride(dim)].view(
            rows, int8_row_stride(dim)
        )

    @torch.no_grad()
    def forward(
        self,
        indices: torch.Tensor,
        offsets: torch.Tensor,
        per_sample_weights: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        B = (offsets.numel() - 1) // self._num_features
        if self.qweights.is_cuda or (self.qweights.is_pinned() and indices.is_cuda):
            ops.hip_ops()
            return torch.ops.trec_amd.tbe_forward_pooled_int8(
                self.qweights,
                self._table_byte_offsets,
                self._dims_t,
                self._feat_table_t,
                self._d_out_offsets,
                indices,
                offsets,
                per_sample_weights if per_sample_weights is not None else self._empty_f,
                B,
                self._total_D,
                self._max_D,
                self.pooling == PoolingType.MEAN,
            )
        # CPU reference
        outs = []
        for f, t in enumerate(self._feature_table_map):
            name, rows, dim = self._specs[t]
            w = dequantize_rowwise_int8(self.packed_table(t), dim)
            off = offsets[f * B : (f + 1) * B + 1] - offsets[f * B]
            idx = indices[int(offsets[f * B]) : int(offsets[(f + 1) * B])]
            pw = (
                per_sample_weights[int(offsets[f * B]) : int(offsets[(f + 1) * B])]
                if per_sample_weights is not None
                else None
            )
            outs.append(
                torch.nn.functional.embedding_bag(
                    idx,
                    w,
                    off,
                    mode="mean" if self.pooling == PoolingType.MEAN else "sum",
                    per_sample_weights=pw,
                    include_last_offset=True,
                )
            )
        return torch.cat(outs, dim=1)


class EmbeddingBagCollection(nn.Module):
    """Quantized EBC (reference quant/embedding_modules.py:346)."""

    def __init__(
        self,
        tables: List[EmbeddingBagConfig],
        is_weighted: bool = False,
        device: Optional[torch.device] = None,
        output_dtype: torch.dtype = torch.float32,
    ) -> None:
        super().__init__()
        self._embedding_bag_configs = tables
        self._is_weighted = is_weighted
        self._feature_names = [f for t in tables for f in t.feature_names]
        self._lengths_per_embedding = [t.embedding_dim for t in tables for _ in t.feature_names]
        poolings = {t.pooling for t in tables}
        assert len(poolings) <= 1, "uniform pooling per quant EBC group"
        self._tbe = QuantTableBatchedEmbeddingBags(
            [(t.name, t.num_embeddings, t.embedding_dim) for t in tables],
            feature_table_map=[i for i, t in enumerate(tables) for _ in t.feature_names],
            pooling=next(iter(poolings)) if poolings else PoolingType.SUM,
            device=device,
        )

    @classmethod
    def from_float(cls, module: FloatEBC, output_dtype: torch.dtype = torch.float32):
        tables = module.embedding_bag_configs()
        device = next(module.parameters()).device if any(True for _ in module.parameters()) else torch.device("cpu")
        q = cls(
            tables=[
                EmbeddingBagConfig(
                    num_embeddings=t.num_embeddings,
                    embedding_dim=t.embedding_dim,
                    name=t.name,
                    feature_names=list(t.feature_names),
                    pooling=t.pooling,
                    data_type=DataType.INT8,
                )
                for t in tables
            ],
            is_weighted=module.is_weighted(),
            device=device,
        )
        for i, t in enumerate(tables):
            q._tbe.load_float_table(i, module.embedding_bags[t.name].weight.detach())
        return q

    def embedding_bag_configs(self) -> List[EmbeddingBagConfig]:
        return self._embedding_bag_configs

    def is_weighted(self) -> bool:
        return self._is_weighted

    def forward(self, features: KeyedJaggedTensor) -> KeyedTensor:
        if features.keys() != self._feature_names:
            order = [features.keys().index(f) for f in self._feature_names]
            features = features.permute(order)
        values = self._tbe(
            features.values(),
            features.offsets(),
            features.weights_or_none() if self._is_weighted else None,
        )
        return KeyedTensor(
            keys=self._feature_names,
            values=values,
            length_per_key=self._lengths_per_embedding,
        )


class EmbeddingCollection(nn.Module):
    """Quantized sequence EC (reference quant/embedding_modules.py:748)."""

    def __init__(
        self,
        tables: List[EmbeddingConfig],
        device: Optional[torch.device] = None,
        output_dtype: torch.dtype = torch.float32,
    ) -> None:
        super().__init__()
        self._embedding_configs = tables
        self._feature_names = [f for t in tables for f in t.feature_names]
        dims = {t.embedding_dim for t in tables}
        assert len(dims) <= 1
        self._dim = next(iter(dims)) if dims else 0
        self._tbe = QuantTableBatchedEmbeddingBags(
            [(t.name, t.num_embeddings, t.embedding_dim) for t in tables],
            feature_table_map=[i for i, t in enumerate(tables) for _ in t.feature_names],
            device=device,
        )

    @classmethod
    def from_float(cls, module: FloatEC, output_dtype: torch.dtype = torch.float32):
        tables = module.embedding_configs()
        q = cls(tables=tables)
        for i, t in enumerate(tables):
            q._tbe.load_float_table(i, module.embeddings[t.name].weight.detach())
        return q

    def embedding_configs(self) -> List[EmbeddingConfig]:
        return self._embedding_configs

    @torch.no_grad()
    def forward(self, features: KeyedJaggedTensor) -> Dict[str, JaggedTensor]:
        if features.keys() != self._feature_names:
            order = [features.keys().index(f) for f in self._feature_names]
            features = features.permute(order)
        tbe = self._tbe
        F = tbe._num_features
        B = features.stride()
        offsets = features.offsets()
        if tbe.qweights.is_cuda or (
            tbe.qweights.is_pinned() and features.values().is_cuda
        ):
            ops.hip_ops()
            feat_val_offsets = offsets[:: B][: F + 1].contiguous()
            rows = torch.ops.trec_amd.tbe_forward_seq_int8(
                tbe.qweights,
                tbe._table_byte_offsets,
                tbe._dims_t,
                tbe._feat_table_t,
                feat_val_offsets,
                features.values(),
                self._dim,
                tbe._max_D,
            )
        else:
            rows_l = []
            for f, t in enumerate(tbe._feature_table_map):
                name, r, dim = tbe._specs[t]
                w = dequantize_rowwise_int8(tbe.packed_table(t), dim)
                idx = features.values()[int(offsets[f * B]) : int(offsets[(f + 1) * B])]
                rows_l.append(w[idx])
            rows = torch.cat(rows_l, dim=0) if rows_l else torch.empty(0, self._dim)
        out: Dict[str, JaggedTensor] = {}
        opk = features.offset_per_key()
        lengths = features.lengths()
        for i, f in enumerate(self._feature_names):
            out[f] = JaggedTensor(
                values=rows[opk[i] : opk[i + 1]],
                lengths=lengths[i * B : (i + 1) * B],
            )
        return out
