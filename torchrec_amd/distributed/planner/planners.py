"""Sharding planner: enumerate -> propose -> partition -> score -> best.

Reference parity: torchrec/distributed/planner/ — EmbeddingShardingPlanner
(planners.py:668), EmbeddingEnumerator (enumerators.py:81), proposers
(proposers.py:34,137), GreedyPerfPartitioner (partitioners.py:176), perf /
storage estimation (shard_estimators.py:71,126). Condensed into one module;
the cost model uses the MI355X constants (constants.py).
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional, Tuple, cast

import torch
import torch.distributed as dist
import torch.nn as nn

from torchrec_amd.distributed.planner import constants
from torchrec_amd.distributed.planner.types import (
    DeviceHardware,
    ParameterConstraints,
    Perf,
    PlannerError,
    Shard,
    ShardingOption,
    Storage,
    Topology,
)
from torchrec_amd.distributed.types import (
    EmbeddingComputeKernel,
    EmbeddingModuleShardingPlan,
    ModuleSharder,
    ParameterSharding,
    ShardMetadata,
    ShardingPlan,
    ShardingPlanner,
    ShardingType,
)

logger = logging.getLogger(__name__)

OPTIMIZER_MULTIPLIER = {"rowwise_adagrad": 1.0 / 128, "sgd": 0.0, "adagrad": 1.0}


def _shard_bytes(rows: int, cols: int, elem: int = 4, opt_mult: float = 1.0 / 128) -> int:
    return int(rows * cols * elem * (1 + opt_mult))


def calculate_shards(
    rows: int,
    cols: int,
    sharding_type: str,
    world_size: int,
    min_partition: int = 64,
    local_size: int = 8,
) -> List[Tuple[List[int], List[int]]]:
    """(sizes, offsets) per shard (reference sharding_plan.py:94)."""
    if sharding_type in (ShardingType.TABLE_WISE.value, ShardingType.DATA_PARALLEL.value):
        n = 1 if sharding_type == ShardingType.TABLE_WISE.value else world_size
        return [([rows, cols], [0, 0]) for _ in range(n)]
    if sharding_type == ShardingType.GRID_SHARD.value:
        from torchrec_amd.distributed.sharding.cw_sharding import cw_shard_dims

        L = min(local_size, world_size)
        NN = max(1, world_size // L)
        col_dims = cw_shard_dims(cols, NN)
        out = []
        for n in range(NN):
            col_off = sum(col_dims[:n])
            block = (rows + L - 1) // L
            for r in range(L):
                lo = min(r * block, rows)
                hi = min((r + 1) * block, rows)
                out.append(([hi - lo, col_dims[n]], [lo, col_off]))
        return out
    if sharding_type == ShardingType.TABLE_ROW_WISE.value:
        L = min(local_size, world_size)
        block = (rows + L - 1) // L
        out = []
        for r in range(L):
            lo = min(r * block, rows)
            hi = min((r + 1) * block, rows)
            out.append(([hi - lo, cols], [lo, 0]))
        return out
    if sharding_type == ShardingType.ROW_WISE.value:
        block = (rows + world_size - 1) // world_size
        out = []
        for r in range(world_size):
            lo = min(r * block, rows)
            hi = min((r + 1) * block, rows)
            out.append(([hi - lo, cols], [lo, 0]))
        return out
    if sharding_type in (
        ShardingType.COLUMN_WISE.value,
        ShardingType.TABLE_COLUMN_WISE.value,
    ):
        # split into shards of >= min_partition cols, at most world_size shards
        # (TWCW: at most local_size shards so the whole table fits one node)
        cap = min(local_size, world_size) if (
            sharding_type == ShardingType.TABLE_COLUMN_WISE.value
        ) else world_size
        n_shards = min(cap, max(1, cols // max(4, min_partition)))
        base = cols // n_shards
        base -= base % 4
        out = []
        off = 0
        for i in range(n_shards):
            w = cols - off if i == n_shards - 1 else base
            out.append(([rows, w], [0, off]))
            off += w
        return out
    raise PlannerError(f"unsupported sharding type {sharding_type}")


class EmbeddingEnumerator:
    """Cross product table x sharding_type x kernel (reference enumerators.py:81)."""

    def __init__(
        self,
        topology: Topology,
        constraints: Optional[Dict[str, ParameterConstraints]] = None,
    ) -> None:
        self._topology = topology
        self._constraints = constraints or {}

    def enumerate(
        self, module: nn.Module, sharders: List[ModuleSharder[nn.Module]]
    ) -> List[ShardingOption]:
        options: List[ShardingOption] = []
        W = self._topology.world_size
        sharder_by_type = {s.module_type: s for s in sharders}
        for fqn, child in module.named_modules():
            sharder = sharder_by_type.get(type(child))
            if sharder is None:
                continue
            if getattr(sharder, "plan_optional", False):
                continue  # e.g. tower sharders place modules themselves
            configs = child.embedding_bag_configs() if hasattr(child, "embedding_bag_configs") else child.embedding_configs()
            is_weighted = child.is_weighted() if hasattr(child, "is_weighted") else False
            for cfg in configs:
                cons = self._constraints.get(cfg.name)
                allowed_types = (
                    cons.sharding_types
                    if cons and cons.sharding_types
                    else sharder.sharding_types(self._topology.compute_device)
                )
                for st in allowed_types:
                    if W == 1 and st not in (
                        ShardingType.TABLE_WISE.value,
                        ShardingType.DATA_PARALLEL.value,
                    ):
                        continue
                    kernels = (
                        cons.compute_kernels
                        if cons and cons.compute_kernels
                        else sharder.compute_kernels(st, self._topology.compute_device)
                    )
                    if (
                        self._topology.compute_device == "cuda"
                        and st != ShardingType.DATA_PARALLEL.value
                        and EmbeddingComputeKernel.FUSED.value in kernels
                        and EmbeddingComputeKernel.FUSED_UVM.value not in kernels
                    ):
                        # host-DRAM spill variants for tables beyond HBM
                        kernels = kernels + [
                            EmbeddingComputeKernel.FUSED_UVM_CACHING.value,
                            EmbeddingComputeKernel.FUSED_UVM.value,
                        ]
                    for kernel in kernels:
                        shards = [
                            Shard(size=list(sz), offset=list(off))
                            for sz, off in calculate_shards(
                                cfg.num_embeddings,
                                cfg.embedding_dim,
                                st,
                                W,
                                min_partition=(cons.min_partition if cons and cons.min_partition else 64),
                                local_size=self._topology.local_world_size,
                            )
                        ]
                        opt = ShardingOption(
                            name=cfg.name,
                            module_fqn=fqn,
                            config=cfg,
                            sharding_type=st,
                            compute_kernel=kernel,
                            shards=shards,
                            is_weighted=is_weighted,
                        )
                        self._estimate(opt, cons)
                        options.append(opt)
        return options

    def _estimate(self, opt: ShardingOption, cons: Optional[ParameterConstraints]) -> None:
        """Perf + storage per shard with per-sharding-type I/O models
        (reference shard_estimators.py:71 perf, :126 storage, :702-862
        _calculate_{tw,rw,cw,twrw}_shard_io_sizes — re-derived for MI355X:
        xGMI collectives are per-link bound, so a2a/RS/AG costs carry the
        (W-1)/W ring factor against intra_host_bw)."""
        topo = self._topology
        B = topo.batch_size
        W = topo.world_size
        L = max(1, topo.local_world_size)
        pooling = cons.pooling_factors[0] if cons else constants.POOLING_FACTOR
        D = opt.config.embedding_dim
        elem = 4  # fp32
        id_bytes = 8  # int64 ids on the input-dist wire
        opt_mult = (
            0.0
            if opt.compute_kernel == EmbeddingComputeKernel.DENSE.value
            else 1.0 / D  # rowwise adagrad: one momentum scalar per row
        )
        is_uvm = opt.compute_kernel == EmbeddingComputeKernel.FUSED_UVM.value
        is_cached = opt.compute_kernel == EmbeddingComputeKernel.FUSED_UVM_CACHING.value
        CACHE_LOAD = 0.2
        CACHE_HIT = 0.8  # assumed; refine with table stats
        n_feat = max(1, len(opt.config.feature_names))
        ring = (W - 1) / W if W > 1 else 0.0
        ring_node = (max(1, W // L) - 1) / max(1, W // L)
        st = opt.sharding_type
        for shard in opt.shards:
            rows, cols = shard.size
            # storage: weights + optimizer + a slice of activation/grad buffers
            weight_bytes = int(rows * cols * elem * (1 + opt_mult))
            act_bytes = int(B * W * cols * elem * 4)
            if is_uvm:
                shard.storage = Storage(hbm=act_bytes, ddr=weight_bytes)
            elif is_cached:
                cache_bytes = int(weight_bytes * CACHE_LOAD)
                shard.storage = Storage(hbm=act_bytes + cache_bytes, ddr=weight_bytes)
            else:
                shard.storage = Storage(hbm=weight_bytes + act_bytes, ddr=0)

            # ---- per-sharding-type I/O (bytes on this shard's rank) ----
            prefetch = 0.0
            if st == ShardingType.DATA_PARALLEL.value:
                # local batch through a DENSE autograd kernel + grad allreduce
                fwd_bytes = (
                    B * pooling * n_feat * cols * elem
                ) * constants.DP_ELEMENTWISE_KERNELS_PERF_FACTOR
                input_comms = 0.0
                output_comms = 0.0
                bwd_comms = (
                    2.0 * ring * rows * cols * elem / topo.intra_host_bw
                    if W > 1 else 0.0
                )
            elif st == ShardingType.ROW_WISE.value:
                # every rank reads ~1/W of the GLOBAL batch's ids; pooled
                # partials reduce-scatter back (reference
                # _calculate_rw_shard_io_sizes)
                gB = B * W
                fwd_bytes = gB * pooling * n_feat / W * cols * elem
                input_comms = gB * pooling * n_feat / W * id_bytes / topo.intra_host_bw
                output_comms = (
                    ring * gB * n_feat * cols * elem / topo.intra_host_bw
                )
                bwd_comms = output_comms  # AG of grads mirrors the RS
            elif st == ShardingType.TABLE_ROW_WISE.value:
                # rows split over one node: intra-node RS + cross-node a2a
                gB = B * W
                fwd_bytes = gB * pooling * n_feat / L * cols * elem
                input_comms = gB * pooling * n_feat / L * id_bytes / topo.intra_host_bw
                rs = (L - 1) / L * gB * n_feat * cols * elem / topo.intra_host_bw
                a2a = ring_node * gB * n_feat * cols * elem / topo.inter_host_bw
                output_comms = rs + a2a
                bwd_comms = output_comms
            elif st == ShardingType.GRID_SHARD.value:
                gB = B * W
                fwd_bytes = gB * pooling * n_feat / L * cols * elem
                input_comms = gB * pooling * n_feat / L * id_bytes / topo.intra_host_bw
                rs = (L - 1) / L * gB * n_feat * cols * elem / topo.intra_host_bw
                a2a = ring_node * gB * n_feat * cols * elem / topo.inter_host_bw
                output_comms = rs + a2a
                bwd_comms = output_comms
            else:
                # TW / CW / TWCW: the GLOBAL batch's bags flow through this
                # shard; ids a2a in, pooled slice a2a out (reference
                # _calculate_tw/cw_shard_io_sizes)
                gB = B * W
                fwd_bytes = gB * pooling * n_feat * cols * elem
                input_comms = (
                    ring * gB * pooling * n_feat * id_bytes / topo.intra_host_bw
                )
                output_comms = (
                    ring * gB * n_feat * cols * elem / topo.intra_host_bw
                )
                bwd_comms = output_comms
            if W == 1:
                input_comms = output_comms = bwd_comms = 0.0

            if is_uvm:
                mem_bw = topo.ddr_mem_bw
            elif is_cached:
                mem_bw = 1.0 / (CACHE_HIT / topo.hbm_mem_bw + (1 - CACHE_HIT) / topo.ddr_mem_bw)
                # cache maintenance traffic ahead of the forward
                prefetch = (1 - CACHE_HIT) * fwd_bytes / topo.ddr_mem_bw
            else:
                mem_bw = topo.hbm_mem_bw
            # narrow-shard penalty (CW/TWCW/GRID): the TBE assigns 64/LPS
            # bags per 64-lane wavefront with LPS lanes covering D/4 float4
            # columns — shards narrower than 64 columns leave lanes idle
            # (D=16 -> 4 of 16 lanes active), so their effective bandwidth
            # shrinks proportionally
            lane_eff = min(1.0, cols / 64.0) if topo.compute_device == "cuda" else 1.0
            fwd_compute = fwd_bytes / (mem_bw * max(lane_eff, 1e-3))
            shard.perf = Perf(
                fwd_compute=fwd_compute,
                fwd_comms=input_comms + output_comms,
                bwd_compute=fwd_compute * constants.BWD_COMPUTE_MULTIPLIER,
                bwd_comms=bwd_comms,
                prefetch_compute=prefetch,
            )


class GreedyPerfPartitioner:
    """Bin-pack shards onto devices by perf (reference partitioners.py:176)."""

    def partition(
        self, proposal: List[ShardingOption], topology: Topology
    ) -> List[ShardingOption]:
        devices = [
            DeviceHardware(d.rank, Storage(d.storage.hbm, d.storage.ddr), Perf())
            for d in topology.devices
        ]
        # fixed-rank types first (RW/DP span all devices)
        for opt in proposal:
            if opt.sharding_type in (
                ShardingType.ROW_WISE.value,
                ShardingType.DATA_PARALLEL.value,
                ShardingType.GRID_SHARD.value,
            ):
                for r, shard in enumerate(opt.shards):
                    shard.rank = r
                    dev = devices[r]
                    dev.storage = dev.storage - shard.storage
                    dev.perf = dev.perf + shard.perf
                    if dev.storage.hbm < 0:
                        raise PlannerError(f"OOM on rank {r} for {opt.name}")
        # TWRW/TWCW: place the whole shard group on one node (greedy by load)
        L = topology.local_world_size
        for opt in proposal:
            if opt.sharding_type not in (
                ShardingType.TABLE_ROW_WISE.value,
                ShardingType.TABLE_COLUMN_WISE.value,
            ):
                continue
            n_nodes = max(1, topology.world_size // L)
            best_node, best_load = None, None
            for node in range(n_nodes):
                devs = devices[node * L : node * L + len(opt.shards)]
                if all(s.storage.fits_in(d.storage) for s, d in zip(opt.shards, devs)):
                    load = max(d.perf.total for d in devs)
                    if best_load is None or load < best_load:
                        best_node, best_load = node, load
            if best_node is None:
                raise PlannerError(f"no node fits TWRW table {opt.name}")
            for i, shard in enumerate(opt.shards):
                dev = devices[best_node * L + i]
                shard.rank = dev.rank
                dev.storage = dev.storage - shard.storage
                dev.perf = dev.perf + shard.perf
        # greedy for TW/CW: biggest perf first onto least-loaded feasible device
        movable = [
            (shard, opt)
            for opt in proposal
            if opt.sharding_type
            in (ShardingType.TABLE_WISE.value, ShardingType.COLUMN_WISE.value)
            for shard in opt.shards
        ]
        movable.sort(key=lambda x: x[0].perf.total if x[0].perf else 0, reverse=True)
        for shard, opt in movable:
            feasible = [d for d in devices if shard.storage.fits_in(d.storage)]
            if not feasible:
                raise PlannerError(f"no device fits shard of {opt.name}")
            dev = min(feasible, key=lambda d: d.perf.total)
            shard.rank = dev.rank
            dev.storage = dev.storage - shard.storage
            dev.perf = dev.perf + shard.perf
        return proposal


class MemoryBalancedPartitioner(GreedyPerfPartitioner):
    """Balance HBM instead of perf for the movable (TW/CW) shards
    (reference partitioners.py:694): picks the feasible device with the most
    free HBM. Used when capacity, not step time, is the binding constraint."""

    def partition(self, proposal, topology):
        # fixed-rank and node-local placements follow the perf partitioner
        placed = super().partition(proposal, topology)
        # rebalance movable shards by storage: biggest first onto emptiest
        devices = {d.rank: [topology.devices[d.rank].storage.hbm, 0.0] for d in topology.devices}
        for opt in placed:
            for shard in opt.shards:
                if opt.sharding_type in (
                    ShardingType.TABLE_WISE.value,
                    ShardingType.COLUMN_WISE.value,
                ):
                    continue
                devices[shard.rank][1] += shard.storage.hbm if shard.storage else 0
        movable = [
            (shard, opt)
            for opt in placed
            if opt.sharding_type
            in (ShardingType.TABLE_WISE.value, ShardingType.COLUMN_WISE.value)
            for shard in opt.shards
        ]
        movable.sort(key=lambda x: x[0].storage.hbm if x[0].storage else 0, reverse=True)
        for shard, opt in movable:
            need = shard.storage.hbm if shard.storage else 0
            feasible = [r for r, (cap, used) in devices.items() if cap - used >= need]
            if not feasible:
                raise PlannerError(f"no device fits shard of {opt.name}")
            r = max(feasible, key=lambda r: devices[r][0] - devices[r][1])
            shard.rank = r
            devices[r][1] += need
        return placed


class GreedyProposer:
    """Per-table best-perf choice, tables in size-desc order
    (reference proposers.py:34)."""

    def propose(self, options: List[ShardingOption]) -> List[List[ShardingOption]]:
        by_table: Dict[Tuple[str, str], List[ShardingOption]] = {}
        for o in options:
            by_table.setdefault((o.module_fqn, o.name), []).append(o)
        proposals: List[List[ShardingOption]] = []
        # proposal k: per table, k-th best option by estimated perf
        max_k = max(len(v) for v in by_table.values()) if by_table else 0
        for k in range(min(max_k, 4)):
            prop = []
            for opts in by_table.values():
                ranked = sorted(opts, key=lambda o: o.total_perf)
                prop.append(ranked[min(k, len(ranked) - 1)])
            proposals.append([self._clone(o) for o in prop])
        return proposals

    @staticmethod
    def _clone(o: ShardingOption) -> ShardingOption:
        return ShardingOption(
            name=o.name,
            module_fqn=o.module_fqn,
            config=o.config,
            sharding_type=o.sharding_type,
            compute_kernel=o.compute_kernel,
            shards=[
                Shard(list(s.size), list(s.offset), None, s.storage, s.perf)
                for s in o.shards
            ],
            is_weighted=o.is_weighted,
        )


class UniformProposer(GreedyProposer):
    """All tables use the same sharding type (reference proposers.py:137)."""

    def propose(self, options: List[ShardingOption]) -> List[List[ShardingOption]]:
        proposals = []
        for st in [
            ShardingType.TABLE_WISE.value,
            ShardingType.ROW_WISE.value,
            ShardingType.COLUMN_WISE.value,
            ShardingType.DATA_PARALLEL.value,
        ]:
            by_table: Dict[Tuple[str, str], ShardingOption] = {}
            ok = True
            for o in options:
                if o.sharding_type != st:
                    continue
                key = (o.module_fqn, o.name)
                if key not in by_table or o.total_perf < by_table[key].total_perf:
                    by_table[key] = o
            n_tables = len({(o.module_fqn, o.name) for o in options})
            if by_table and len(by_table) == n_tables:
                proposals.append([self._clone(o) for o in by_table.values()])
        return proposals


class GridSearchProposer(GreedyProposer):
    """Exhaustive cross product of per-table options — only viable for small
    models (reference proposers.py:207 caps the same way)."""

    MAX_PROPOSALS = 256

    def propose(self, options: List[ShardingOption]) -> List[List[ShardingOption]]:
        import itertools

        by_table: Dict[Tuple[str, str], List[ShardingOption]] = {}
        for o in options:
            by_table.setdefault((o.module_fqn, o.name), []).append(o)
        total = 1
        for opts in by_table.values():
            total *= len(opts)
            if total > self.MAX_PROPOSALS:
                return []  # defer to the other proposers
        return [
            [self._clone(o) for o in combo]
            for combo in itertools.product(*by_table.values())
        ]


class DynamicProgrammingProposer(GreedyProposer):
    """HBM-budgeted DP over tables (reference proposers.py:287): tables are
    stages, HBM is discretized into bins, and dp[t][h] holds the min total
    estimated perf using <= h HBM for the first t tables. Finds plans the
    per-table greedy misses when the best-perf option of a big table starves
    smaller ones out of HBM."""

    BINS = 64

    def propose(self, options: List[ShardingOption]) -> List[List[ShardingOption]]:
        by_table: Dict[Tuple[str, str], List[ShardingOption]] = {}
        for o in options:
            by_table.setdefault((o.module_fqn, o.name), []).append(o)
        if not by_table:
            return []
        tables = list(by_table.values())
        total_hbm = sum(
            max(sum(s.storage.hbm for s in o.shards) for o in opts)
            for opts in tables
        )
        if total_hbm <= 0:
            return []
        bin_sz = max(1, total_hbm // self.BINS)
        B = self.BINS + 1
        INF = float("inf")
        dp = [[INF] * B for _ in range(len(tables) + 1)]
        choice: List[List[Optional[int]]] = [[None] * B for _ in range(len(tables))]
        for h in range(B):
            dp[0][h] = 0.0
        for t, opts in enumerate(tables):
            for h in range(B):
                if dp[t][h] == INF:
                    continue
                for oi, o in enumerate(opts):
                    hbm = sum(s.storage.hbm for s in o.shards)
                    nh = min(B - 1, h + max(1, int(hbm // bin_sz)))
                    cost = dp[t][h] + o.total_perf
                    if cost < dp[t + 1][nh]:
                        dp[t + 1][nh] = cost
                        choice[t][nh] = (h, oi)
        # backtrack the best terminal bin
        best_h = min(range(B), key=lambda h: dp[len(tables)][h])
        if dp[len(tables)][best_h] == INF:
            return []
        prop: List[ShardingOption] = []
        h = best_h
        for t in range(len(tables) - 1, -1, -1):
            back = choice[t][h]
            if back is None:
                return []
            h, oi = back
            prop.append(self._clone(tables[t][oi]))
        prop.reverse()
        return [prop]


class UVMSpillProposer(GreedyProposer):
    """Best DEVICE-kernel plan, spilling the k largest tables to host DRAM
    (FUSED_UVM) — covers models beyond 288 GB HBM (reference
    EmbeddingOffloadScaleupProposer role, proposers.py:471)."""

    def propose(self, options: List[ShardingOption]) -> List[List[ShardingOption]]:
        by_table: Dict[Tuple[str, str], List[ShardingOption]] = {}
        for o in options:
            by_table.setdefault((o.module_fqn, o.name), []).append(o)
        # order tables by size desc
        sizes = {
            k: max(o.config.num_embeddings * o.config.embedding_dim for o in v)
            for k, v in by_table.items()
        }
        ordered = sorted(by_table.keys(), key=lambda k: -sizes[k])
        proposals = []
        k = 1
        while k <= len(ordered):
            spill = set(ordered[:k])
            prop = []
            ok = True
            for key, opts in by_table.items():
                want_uvm = key in spill
                cands = [
                    o
                    for o in opts
                    if (o.compute_kernel == EmbeddingComputeKernel.FUSED_UVM.value)
                    == want_uvm
                ]
                if not cands:
                    ok = False
                    break
                prop.append(min(cands, key=lambda o: o.total_perf))
            if ok:
                proposals.append([self._clone(o) for o in prop])
            k *= 2
        return proposals


class EmbeddingShardingPlanner(ShardingPlanner):
    """Reference parity: planners.py:668."""

    def __init__(
        self,
        topology: Optional[Topology] = None,
        constraints: Optional[Dict[str, ParameterConstraints]] = None,
        batch_size: Optional[int] = None,
        storage_reservation=None,
    ) -> None:
        self._storage_reservation = storage_reservation
        if topology is None:
            topology = Topology(
                world_size=dist.get_world_size() if dist.is_initialized() else 1,
                compute_device="cuda" if torch.cuda.is_available() else "cpu",
            )
        if batch_size is not None:
            topology.batch_size = batch_size
        self._topology = topology
        self._enumerator = EmbeddingEnumerator(topology, constraints)
        self._partitioner = GreedyPerfPartitioner()
        self._proposers = [
            GreedyProposer(),
            UniformProposer(),
            GridSearchProposer(),
            DynamicProgrammingProposer(),
            UVMSpillProposer(),
        ]

    def plan(
        self, module: nn.Module, sharders: List[ModuleSharder[nn.Module]]
    ) -> ShardingPlan:
        if self._storage_reservation is not None:
            self._topology = self._storage_reservation.reserve(
                self._topology, module, self._topology.batch_size
            )
        options = self._enumerator.enumerate(module, sharders)
        if not options:
            return ShardingPlan({})
        best: Optional[List[ShardingOption]] = None
        best_score = float("inf")
        errors = []
        for proposer in self._proposers:
            for proposal in proposer.propose(options):
                try:
                    placed = self._partitioner.partition(proposal, self._topology)
                except PlannerError as e:
                    errors.append(str(e))
                    continue
                # score = max per-device perf (critical path)
                per_dev = [0.0] * self._topology.world_size
                for opt in placed:
                    for shard in opt.shards:
                        per_dev[shard.rank] += shard.perf.total if shard.perf else 0
                score = max(per_dev)
                if score < best_score:
                    best_score = score
                    best = placed
        if best is None:
            raise PlannerError(
                f"no feasible sharding plan found; partition errors: {errors[:3]}"
            )
        from torchrec_amd.distributed.planner.stats import EmbeddingStats

        self.last_stats = EmbeddingStats().log(best, self._topology)
        return self._to_sharding_plan(best)

    def collective_plan(
        self, module: nn.Module, sharders: List[ModuleSharder[nn.Module]], pg
    ) -> ShardingPlan:
        """Plan on rank 0, broadcast (reference planners.py collective_plan)."""
        if pg is None or dist.get_world_size(pg) == 1:
            return self.plan(module, sharders)
        if dist.get_rank(pg) == 0:
            plan = self.plan(module, sharders)
            obj = [plan]
        else:
            obj = [None]
        dist.broadcast_object_list(obj, src=0, group=pg)
        return obj[0]

    def _to_sharding_plan(self, options: List[ShardingOption]) -> ShardingPlan:
        plan: Dict[str, EmbeddingModuleShardingPlan] = {}
        for opt in options:
            mplan = plan.setdefault(opt.module_fqn, EmbeddingModuleShardingPlan())
            mplan.plan[opt.name] = ParameterSharding(
                sharding_type=opt.sharding_type,
                compute_kernel=opt.compute_kernel,
                ranks=[s.rank for s in opt.shards],
                sharding_spec=[
                    ShardMetadata(
                        shard_offsets=list(s.offset),
                        shard_sizes=list(s.size),
                        placement_rank=s.rank,
                    )
                    for s in opt.shards
                ],
            )
        return ShardingPlan(plan)
