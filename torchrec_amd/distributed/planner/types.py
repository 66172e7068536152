"""Planner data model.

Reference parity: torchrec/distributed/planner/types.py (Topology :986,
Storage/Perf, ShardingOption :1299, ParameterConstraints :1586).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from torchrec_amd.distributed.planner import constants
from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig


@dataclass
class Storage:
    hbm: int = 0
    ddr: int = 0

    def __add__(self, other: "Storage") -> "Storage":
        return Storage(self.hbm + other.hbm, self.ddr + other.ddr)

    def __sub__(self, other: "Storage") -> "Storage":
        return Storage(self.hbm - other.hbm, self.ddr - other.ddr)

    def fits_in(self, other: "Storage") -> bool:
        return self.hbm <= other.hbm and self.ddr <= other.ddr


@dataclass
class Perf:
    """Estimated per-step cost (seconds) of one shard (reference types.py)."""

    fwd_compute: float = 0.0
    fwd_comms: float = 0.0
    bwd_compute: float = 0.0
    bwd_comms: float = 0.0
    prefetch_compute: float = 0.0

    @property
    def total(self) -> float:
        return (
            self.fwd_compute + self.fwd_comms + self.bwd_compute + self.bwd_comms
            + self.prefetch_compute
        )

    def __add__(self, other: "Perf") -> "Perf":
        return Perf(
            self.fwd_compute + other.fwd_compute,
            self.fwd_comms + other.fwd_comms,
            self.bwd_compute + other.bwd_compute,
            self.bwd_comms + other.bwd_comms,
            self.prefetch_compute + other.prefetch_compute,
        )


class DeviceHardware:
    def __init__(self, rank: int, storage: Storage, perf: Perf) -> None:
        self.rank = rank
        self.storage = storage
        self.perf = perf


class Topology:
    """Cluster hardware model (reference planner/types.py:986) with MI355X
    defaults from planner/constants.py."""

    def __init__(
        self,
        world_size: int,
        compute_device: str = "cuda",
        hbm_cap: Optional[int] = None,
        ddr_cap: Optional[int] = None,
        local_world_size: Optional[int] = None,
        hbm_mem_bw: float = constants.HBM_MEM_BW,
        ddr_mem_bw: float = constants.DDR_MEM_BW,
        intra_host_bw: float = constants.INTRA_NODE_BW,
        inter_host_bw: float = constants.INTER_NODE_BW,
        batch_size: int = constants.BATCH_SIZE,
    ) -> None:
        self._world_size = world_size
        self._compute_device = compute_device
        hbm = hbm_cap if hbm_cap is not None else (
            int(constants.HBM_CAP * constants.MAX_HBM_UTILIZATION)
            if compute_device == "cuda"
            else 0
        )
        ddr = ddr_cap if ddr_cap is not None else constants.DDR_CAP
        self._devices = [
            DeviceHardware(r, Storage(hbm=hbm, ddr=ddr), Perf()) for r in range(world_size)
        ]
        self._local_world_size = local_world_size or min(world_size, 8)
        self.hbm_mem_bw = hbm_mem_bw
        self.ddr_mem_bw = ddr_mem_bw
        self.intra_host_bw = intra_host_bw
        self.inter_host_bw = inter_host_bw
        self.batch_size = batch_size

    @property
    def world_size(self) -> int:
        return self._world_size

    @property
    def compute_device(self) -> str:
        return self._compute_device

    @property
    def devices(self) -> List[DeviceHardware]:
        return self._devices

    @property
    def local_world_size(self) -> int:
        return self._local_world_size


@dataclass
class Shard:
    size: List[int]  # [rows, cols]
    offset: List[int]
    rank: Optional[int] = None
    storage: Optional[Storage] = None
    perf: Optional[Perf] = None


@dataclass
class ShardingOption:
    """One candidate (table x sharding_type x kernel) (reference types.py:1299)."""

    name: str
    module_fqn: str
    config: EmbeddingBagConfig
    sharding_type: str
    compute_kernel: str
    shards: List[Shard]
    is_weighted: bool = False

    @property
    def total_storage(self) -> Storage:
        s = Storage()
        for sh in self.shards:
            if sh.storage:
                s = s + sh.storage
        return s

    @property
    def total_perf(self) -> float:
        return sum(sh.perf.total for sh in self.shards if sh.perf)


@dataclass
class ParameterConstraints:
    """Per-table search constraints (reference types.py:1586)."""

    sharding_types: Optional[List[str]] = None
    compute_kernels: Optional[List[str]] = None
    min_partition: Optional[int] = None
    pooling_factors: List[float] = field(default_factory=lambda: [constants.POOLING_FACTOR])


class PlannerError(Exception):
    pass
