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
    MX4 = "mx4"


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
            CommType.MX4: torch.int8,
        }[comm_precision]

    @property
    def precision(self) -> CommType:
        return self._precision

    MX4_GROUP = 32  # values per shared exponent (OCP MX spec)

    def encoded_numel(self, numel: int) -> int:
        if self._precision == CommType.INT8:
            return numel + 4  # + packed fp32 scale
        if self._precision == CommType.MX4:
            assert numel % self.MX4_GROUP == 0, (
                "MX4 wire needs numel % 32 == 0 (use dims that are multiples "
                "of 32 so a2a block splits stay group-aligned)"
            )
            # 32 nibbles -> 16 bytes + 1 shared-exponent byte per group
            return (numel // self.MX4_GROUP) * (self.MX4_GROUP // 2 + 1)
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
        if self._precision == CommType.MX4:
            return self._encode_mx4(t)
        return t.to(self._wire_dtype)

    def _encode_mx4(self, t: torch.Tensor) -> torch.Tensor:
        """OCP MX4-style: groups of 32 share a power-of-two exponent byte;
        each value is a signed 4-bit scaled integer (1 sign + 3 magnitude)."""
        G = self.MX4_GROUP
        v = t.reshape(-1, G).float()
        amax = v.abs().amax(dim=1).clamp(min=1e-30)
        # shared exponent: scale so the max maps to 7
        exp = torch.ceil(torch.log2(amax / 7.0))
        exp = exp.clamp(-127, 127)
        scale = torch.pow(2.0, exp).unsqueeze(1)
        q = (v / scale).round().clamp(-7, 7).to(torch.int8)  # [-7, 7]
        # pack two nibbles per byte (offset-8 so the nibble is unsigned)
        u = (q + 8).to(torch.uint8).reshape(-1, G // 2, 2)
        packed = (u[..., 0] | (u[..., 1] << 4)).reshape(-1, G // 2)
        exp_b = (exp.to(torch.int8) .view(-1, 1)).view(torch.uint8)
        return torch.cat([packed, exp_b], dim=1).reshape(-1).view(torch.int8)

    def decode(self, t: torch.Tensor, numel: int) -> torch.Tensor:
        if self._precision == CommType.FP32:
            return t
        if self._precision == CommType.INT8:
            scale = t[-4:].view(torch.float32)
            return t[:-4].to(torch.float32) * scale
        if self._precision == CommType.FP8:
            return t.view(torch.float8_e4m3fn).to(torch.float32)
        if self._precision == CommType.MX4:
            return self._decode_mx4(t, numel)
        return t.to(torch.float32)

    def _decode_mx4(self, t: torch.Tensor, numel: int) -> torch.Tensor:
        G = self.MX4_GROUP
        rows = numel // G
        b = t.view(torch.uint8).reshape(rows, G // 2 + 1)
        packed = b[:, : G // 2]
        exp = b[:, G // 2].view(torch.int8).to(torch.float32)
        lo = (packed & 0xF).to(torch.int16) - 8
        hi = (packed >> 4).to(torch.int16) - 8
        q = torch.stack([lo, hi], dim=-1).reshape(rows, G).to(torch.float32)
        return (q * torch.pow(2.0, exp).unsqueeze(1)).reshape(-1)


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
