"""Storage reservations: carve dense/KJT/overhead headroom out of HBM before
partitioning (reference: torchrec/distributed/planner/storage_reservations.py
HeuristicalStorageReservation :435, FixedPercentageStorageReservation :198,
InferenceStorageReservation :542)."""

from __future__ import annotations

import abc
from typing import List, Optional

import torch.nn as nn

from torchrec_amd.distributed.planner.types import Topology


class StorageReservation(abc.ABC):
    @abc.abstractmethod
    def reserve(
        self,
        topology: Topology,
        module: Optional[nn.Module] = None,
        batch_size: Optional[int] = None,
    ) -> Topology:
        """Return a topology whose per-device HBM caps exclude the reserve."""


class FixedPercentageStorageReservation(StorageReservation):
    """Reserve a flat fraction of each device's HBM (reference :198)."""

    def __init__(self, percentage: float) -> None:
        assert 0.0 <= percentage < 1.0
        self._pct = percentage

    def reserve(self, topology, module=None, batch_size=None) -> Topology:
        for d in topology.devices:
            d.storage.hbm = int(d.storage.hbm * (1 - self._pct))
        return topology


class HeuristicalStorageReservation(StorageReservation):
    """Reserve dense-parameter + activation/KJT headroom estimated from the
    module (reference :435): dense params (x4 for grads+optimizer) plus a
    per-batch KJT/activation allowance, plus a flat safety fraction."""

    def __init__(self, percentage: float = 0.15, kjt_bytes_per_sample: int = 4096) -> None:
        self._pct = percentage
        self._kjt_bytes = kjt_bytes_per_sample

    def reserve(self, topology, module=None, batch_size=None) -> Topology:
        dense_bytes = 0
        if module is not None:
            for name, p in module.named_parameters():
                if "embedding" in name or "sparse" in name.lower():
                    continue  # sharded tables are what we're budgeting FOR
                dense_bytes += p.numel() * p.element_size()
        B = batch_size or topology.batch_size
        per_device = int(
            dense_bytes * 4  # params + grads + 2x optimizer state
            + B * self._kjt_bytes
        )
        for d in topology.devices:
            d.storage.hbm = int(d.storage.hbm * (1 - self._pct)) - per_device
            if d.storage.hbm < 0:
                d.storage.hbm = 0
        return topology


class InferenceStorageReservation(StorageReservation):
    """Inference: no grads/optimizer — only a flat fraction + dense weights
    (reference :542)."""

    def __init__(self, percentage: float = 0.05) -> None:
        self._pct = percentage

    def reserve(self, topology, module=None, batch_size=None) -> Topology:
        dense_bytes = 0
        if module is not None:
            dense_bytes = sum(
                p.numel() * p.element_size() for p in module.parameters()
            )
        for d in topology.devices:
            d.storage.hbm = int(d.storage.hbm * (1 - self._pct)) - dense_bytes
            if d.storage.hbm < 0:
                d.storage.hbm = 0
        return topology
