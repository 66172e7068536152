"""Planner stats reporting (reference: torchrec/distributed/planner/stats.py
EmbeddingStats :151 — the rich per-table plan table logged after planning)."""

from __future__ import annotations

import logging
from typing import List

logger = logging.getLogger(__name__)


def _fmt_bytes(b: float) -> str:
    for unit in ("B", "KB", "MB", "GB", "TB"):
        if abs(b) < 1024:
            return f"{b:.1f}{unit}"
        b /= 1024
    return f"{b:.1f}PB"


class EmbeddingStats:
    """Renders the chosen plan as an aligned table: per table the sharding
    type, compute kernel, ranks, shard shapes, HBM/DDR bytes and the
    estimated per-shard perf; plus per-device rollups."""

    def log(self, best_options: List, topology) -> str:
        rows = [
            (
                "table",
                "sharding",
                "kernel",
                "ranks",
                "shard shapes",
                "hbm",
                "ddr",
                "perf(ms)",
            )
        ]
        per_dev_hbm = [0.0] * topology.world_size
        per_dev_perf = [0.0] * topology.world_size
        for opt in best_options:
            hbm = sum(s.storage.hbm for s in opt.shards if s.storage)
            ddr = sum(s.storage.ddr for s in opt.shards if s.storage)
            perf = sum(s.perf.total for s in opt.shards if s.perf)
            for s in opt.shards:
                if s.rank is not None and s.storage:
                    per_dev_hbm[s.rank] += s.storage.hbm
                if s.rank is not None and s.perf:
                    per_dev_perf[s.rank] += s.perf.total
            shapes = ",".join(f"{s.size[0]}x{s.size[1]}" for s in opt.shards[:4])
            if len(opt.shards) > 4:
                shapes += f",..x{len(opt.shards)}"
            rows.append(
                (
                    opt.name,
                    opt.sharding_type,
                    opt.compute_kernel,
                    ",".join(str(s.rank) for s in opt.shards[:8]),
                    shapes,
                    _fmt_bytes(hbm),
                    _fmt_bytes(ddr),
                    f"{perf * 1e3:.3f}",
                )
            )
        widths = [max(len(str(r[c])) for r in rows) for c in range(len(rows[0]))]
        lines = []
        for i, r in enumerate(rows):
            lines.append("  ".join(str(v).ljust(w) for v, w in zip(r, widths)))
            if i == 0:
                lines.append("-" * (sum(widths) + 2 * (len(widths) - 1)))
        lines.append("")
        lines.append(
            "per-device HBM: "
            + " ".join(f"r{i}={_fmt_bytes(h)}" for i, h in enumerate(per_dev_hbm))
        )
        lines.append(
            "per-device est perf(ms): "
            + " ".join(f"r{i}={p*1e3:.3f}" for i, p in enumerate(per_dev_perf))
        )
        table = "\n".join(lines)
        logger.info("EmbeddingShardingPlanner stats:\n%s", table)
        return table
