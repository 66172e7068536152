"""Distributed core types.

Reference parity: torchrec/distributed/types.py (ShardingType :142,
ComputeKernel :223, Awaitable :367 / LazyAwaitable :414, ParameterSharding
:770, ShardingPlan :868, ShardingEnv :920, ShardedModule :1200,
ModuleSharder :1409).
"""

from __future__ import annotations

import abc
from dataclasses import dataclass, field
from enum import Enum, unique
from typing import Any, Callable, Dict, Generic, List, Optional, Type, TypeVar

import torch
import torch.distributed as dist
import torch.nn as nn

W = TypeVar("W")
M = TypeVar("M", bound=nn.Module)


@unique
class ShardingType(Enum):
    """How a table's rows/cols are placed (reference types.py:142)."""

    DATA_PARALLEL = "data_parallel"
    TABLE_WISE = "table_wise"
    COLUMN_WISE = "column_wise"
    ROW_WISE = "row_wise"
    TABLE_ROW_WISE = "table_row_wise"
    TABLE_COLUMN_WISE = "table_column_wise"
    GRID_SHARD = "grid_shard"


@unique
class EmbeddingComputeKernel(Enum):
    """Kernel backing a shard (reference embedding_types.py:87)."""

    DENSE = "dense"
    FUSED = "fused"
    FUSED_UVM = "fused_uvm"
    FUSED_UVM_CACHING = "fused_uvm_caching"
    KEY_VALUE = "key_value"  # bounded DRAM virtual table + id translation
    QUANT = "quant"


@dataclass
class CacheParams:
    """UVM-cache tuning knobs riding fused_params (reference types.py:643).

    ``load_factor`` sizes the HBM lxu cache relative to the table;
    ``reserved_memory`` is subtracted from the planner's HBM budget;
    ``precision`` is the cache line dtype (fp32 in v1)."""

    algorithm: str = "lru"
    load_factor: Optional[float] = None
    reserved_memory: Optional[float] = None
    precision: Optional[str] = None
    prefetch_pipeline: Optional[bool] = None


@dataclass
class KeyValueParams:
    """KEY_VALUE (virtual table) kernel knobs (reference types.py:685)."""

    capacity: Optional[int] = None
    eviction: str = "lfu_lru"
    bulk_init_chunk_size: Optional[int] = None


class Awaitable(abc.ABC, Generic[W]):
    """Handle for an async (collective) result (reference types.py:367)."""

    def __init__(self) -> None:
        self._callbacks: List[Callable[[W], W]] = []

    @abc.abstractmethod
    def _wait_impl(self) -> W:
        ...

    def wait(self) -> W:
        ret = self._wait_impl()
        for cb in self._callbacks:
            ret = cb(ret)
        return ret

    @property
    def callbacks(self) -> List[Callable[[W], W]]:
        return self._callbacks


class NoWait(Awaitable[W]):
    def __init__(self, obj: W) -> None:
        super().__init__()
        self._obj = obj

    def _wait_impl(self) -> W:
        return self._obj


class LazyAwaitable(Awaitable[W]):
    """Awaitable that transparently waits on first attribute use.

    Reference parity: types.py:414. Downstream modules can treat the result
    as the real object (e.g. call ``.values()`` on a not-yet-arrived
    KeyedTensor); the first access forces the wait, which lets the collective
    overlap with unrelated compute in between.
    """

    def __init__(self) -> None:
        super().__init__()
        self._result: Optional[W] = None

    def _force(self) -> W:
        if self._result is None:
            self._result = self.wait()
        return self._result

    def __getattr__(self, name: str):
        if name.startswith("_"):
            raise AttributeError(name)
        return getattr(self._force(), name)

    def __getitem__(self, key):
        return self._force()[key]

    def __iter__(self):
        return iter(self._force())

    def __len__(self):
        return len(self._force())

    def __contains__(self, key):
        return key in self._force()


class LazyNoWait(LazyAwaitable[W]):
    def __init__(self, obj: W) -> None:
        super().__init__()
        self._obj = obj

    def _wait_impl(self) -> W:
        return self._obj


@dataclass
class ShardMetadata:
    """Offsets/sizes of one shard of a [rows, dim] table."""

    shard_offsets: List[int]
    shard_sizes: List[int]
    placement_rank: int


@dataclass
class ParameterSharding:
    """Sharding decision for one table (reference types.py:770)."""

    sharding_type: str
    compute_kernel: str
    ranks: Optional[List[int]] = None
    sharding_spec: Optional[List[ShardMetadata]] = None
    cache_params: Optional[Dict[str, Any]] = None


@dataclass
class ModuleShardingPlan:
    pass


@dataclass
class EmbeddingModuleShardingPlan(ModuleShardingPlan):
    """table name -> ParameterSharding (reference types.py:817)."""

    plan: Dict[str, ParameterSharding] = field(default_factory=dict)

    def __getitem__(self, k: str) -> ParameterSharding:
        return self.plan[k]

    def items(self):
        return self.plan.items()

    def values(self):
        return self.plan.values()

    def keys(self):
        return self.plan.keys()

    def __contains__(self, k: str) -> bool:
        return k in self.plan


@dataclass
class ShardingPlan:
    """module FQN -> per-table plan (reference types.py:868)."""

    plan: Dict[str, EmbeddingModuleShardingPlan] = field(default_factory=dict)

    def get_plan_for_module(self, module_path: str) -> Optional[EmbeddingModuleShardingPlan]:
        return self.plan.get(module_path)

    def __str__(self) -> str:
        lines = []
        for fqn, mplan in self.plan.items():
            lines.append(f"module: {fqn}")
            for name, ps in mplan.items():
                lines.append(f"  {name}: {ps.sharding_type}/{ps.compute_kernel} ranks={ps.ranks}")
        return "\n".join(lines)


class ShardingEnv:
    """Wraps the process group (reference types.py:920)."""

    def __init__(self, world_size: int, rank: int, pg: Optional[dist.ProcessGroup] = None) -> None:
        self.world_size = world_size
        self.rank = rank
        self.process_group = pg
        # 2D: rank lists of EVERY sharding group (set by DMPCollection) so
        # per-sharding communicator creation can satisfy new_group's
        # all-ranks-same-arguments collective contract
        self.all_group_ranks: Optional[List[List[int]]] = None

    @classmethod
    def from_process_group(cls, pg: dist.ProcessGroup) -> "ShardingEnv":
        return cls(dist.get_world_size(pg), dist.get_rank(pg), pg)

    @classmethod
    def from_local(cls, world_size: int, rank: int) -> "ShardingEnv":
        """Single-process env (inference / tests)."""
        return cls(world_size, rank, None)


class ShardedModule(abc.ABC, nn.Module, Generic[W]):
    """input_dist / compute / output_dist execution contract
    (reference types.py:1200)."""

    @abc.abstractmethod
    def input_dist(self, ctx, *input, **kwargs) -> Awaitable[Any]:
        ...

    @abc.abstractmethod
    def compute(self, ctx, dist_input) -> Any:
        ...

    @abc.abstractmethod
    def output_dist(self, ctx, output) -> LazyAwaitable[W]:
        ...

    def compute_and_output_dist(self, ctx, dist_input) -> LazyAwaitable[W]:
        return self.output_dist(ctx, self.compute(ctx, dist_input))

    @abc.abstractmethod
    def create_context(self) -> Any:
        ...

    def forward(self, *input, **kwargs) -> LazyAwaitable[W]:
        ctx = self.create_context()
        dist_input = self.input_dist(ctx, *input, **kwargs).wait().wait()
        return self.compute_and_output_dist(ctx, dist_input)


class ModuleSharder(abc.ABC, Generic[M]):
    """Knows how to shard one module type (reference types.py:1409)."""

    @abc.abstractmethod
    def shard(
        self,
        module: M,
        params: EmbeddingModuleShardingPlan,
        env: ShardingEnv,
        device: Optional[torch.device] = None,
    ) -> ShardedModule:
        ...

    @property
    @abc.abstractmethod
    def module_type(self) -> Type[M]:
        ...

    def shardable_parameters(self, module: M) -> Dict[str, nn.Parameter]:
        return dict(module.named_parameters())

    def sharding_types(self, compute_device_type: str) -> List[str]:
        return [
            ShardingType.DATA_PARALLEL.value,
            ShardingType.TABLE_WISE.value,
            ShardingType.COLUMN_WISE.value,
            ShardingType.ROW_WISE.value,
        ]

    def compute_kernels(self, sharding_type: str, compute_device_type: str) -> List[str]:
        if sharding_type == ShardingType.DATA_PARALLEL.value:
            # DP tables must surface dense grads for the DDP allreduce
            return [EmbeddingComputeKernel.DENSE.value]
        return [EmbeddingComputeKernel.FUSED.value]


class ShardingPlanner(abc.ABC):
    """Produces a ShardingPlan (reference types.py:1500)."""

    @abc.abstractmethod
    def plan(self, module: nn.Module, sharders: List[ModuleSharder[nn.Module]]) -> ShardingPlan:
        ...

    @abc.abstractmethod
    def collective_plan(
        self, module: nn.Module, sharders: List[ModuleSharder[nn.Module]], pg
    ) -> ShardingPlan:
        ...
