"""Table-wise column-wise sharding (reference:
torchrec/distributed/sharding/twcw_sharding.py TwCwPooledEmbeddingSharding).

All column shards of a table live on ranks of ONE node, so the fan-out of a
feature's ids and the pooled concat stay on xGMI (intra-node) links. The
dist/lookup machinery is identical to CW — the node-local placement is the
planner's job (GreedyPerfPartitioner places the whole TWCW shard group on the
least-loaded node, mirroring its TWRW path)."""

from __future__ import annotations

from torchrec_amd.distributed.sharding.cw_sharding import CwPooledEmbeddingSharding


class TwCwPooledEmbeddingSharding(CwPooledEmbeddingSharding):
    """CW dists over planner-guaranteed node-local column shards."""
