"""Quantized-communication codecs for pooled/sequence collectives.

Reference parity: torchrec/distributed/fbgemm_qcomm_codec.py (QCommsConfig
:55, precisions FP32/FP16/BF16/FP8/INT8 :31-50) and the QuantizedCommCodec
protocol (torchrec/distributed/types.py:233). The codec plugs into
comm_ops.alltoall_pooled / reduce-scatter via the ``codec`` argument.
"""

from __future__ import annotations

from dataclasses import dataclass
from enum import Enum, unique
from typing import Optional

import torch


@unique
class CommType(Enum):
    FP32 = "fp32"
    FP16 = "fp16"
    BF16 = "bf16"
    FP8 = "fp8"
    INT8 = "int8"


class QuantizedCommCodec:
    """encode/decode a flat fp32 tensor for the wire (reference types.py:233)."""

    def __init__(self, comm_precision: CommType) -> None:
        self._precision = comm_precision
        self._wire_dtype = {
            CommType.FP32: torch.float32,
            CommType.FP16: torch.float16,
            CommType.BF16: torch.bfloat16,
            CommType.FP8: torch.float8_e4m3fn,  # OCP e4m3 (gfx950-native)
            CommType.INT8: torch.int8,
        }[comm_precision]

    @property
    def precision(self) -> CommType:
        return self._precision

    def encoded_numel(self, numel: int) -> int:
        if self._precision == CommType.INT8:
            return numel + 4  # + packed fp32 scale
        return numel

    def encode(self, t: torch.Tensor) -> torch.Tensor:
        if self._precision == CommType.FP32:
            return t
        if self._precision == CommType.INT8:
            # symmetric per-message scale packed into 4 trailing int8 bytes
            scale = (t.abs().max().clamp(min=1e-8) / 127.0).reshape(1)
            q = (t / scale).round().clamp(-127, 127).to(torch.int8)
            return torch.cat([q, scale.view(torch.int8)])
        if self._precision == CommType.FP8:
            # OCP e4m3 payload on an int8 wire (RCCL dtype support)
            return t.to(self._wire_dtype).view(torch.int8)
        return t.to(self._wire_dtype)

    def decode(self, t: torch.Tensor, numel: int) -> torch.Tensor:
        if self._precision == CommType.FP32:
            return t
        if self._precision == CommType.INT8:
            scale = t[-4:].view(torch.float32)
            return t[:-4].to(torch.float32) * scale
        if self._precision == CommType.FP8:
            return t.view(torch.float8_e4m3fn).to(torch.float32)
        return t.to(torch.float32)


@dataclass
class QCommsConfig:
    """Reference parity: fbgemm_qcomm_codec.py:55."""

    forward_precision: CommType = CommType.FP32
    backward_precision: CommType = CommType.FP32


def get_qcomm_codecs(config: Optional[QCommsConfig]):
    if config is None:
        return None, None
    fwd = (
        QuantizedCommCodec(config.forward_precision)
        if config.forward_precision != CommType.FP32
        else None
    )
    bwd = (
        QuantizedCommCodec(config.backward_precision)
        if config.backward_precision != CommType.FP32
        else None
    )
    return fwd, bwd
