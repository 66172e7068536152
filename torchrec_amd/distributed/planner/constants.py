"""Planner hardware-model constants for MI355X (gfx950).

Replaces the reference's A100-class defaults (reference
torchrec/distributed/planner/constants.py:16-45) with MI355X numbers:
HBM3E capacity/bandwidth from /opt/skills/guides/MI355X_MICROARCH.md
(measured: 6.29 TB/s achievable of 8 TB/s peak), xGMI intra-node bandwidth
(7 p2p links x ~153 GB/s per GPU — all-to-all aggregate ~ 1 TB/s egress but
per-ring-link bound for reduce collectives), PCIe Gen5 host link.
"""

# memory capacities (bytes)
HBM_CAP: int = 288 * 1024 * 1024 * 1024  # 288 GB HBM3E
DDR_CAP: int = 1024 * 1024 * 1024 * 1024  # host DRAM assumed 1 TB
DDR_MEM_BW: float = 80 * 1024 * 1024 * 1024  # pinned-host over PCIe Gen5 ~63-80 GB/s

# bandwidths (bytes/s)
HBM_MEM_BW: float = 6.3 * 1024**4 / 1.0995  # ~6.3e12 measured achievable
XGMI_LINK_BW: float = 153 * 1024**3  # one p2p link
INTRA_NODE_BW: float = 7 * XGMI_LINK_BW * 0.8  # a2a across 7 links, 80% eff
INTER_NODE_BW: float = 50 * 1024**3  # 400 Gb/s NIC assumption

CROSS_NODE_BANDWIDTH = INTER_NODE_BW
INTRA_NODE_BANDWIDTH = INTRA_NODE_BW

# compute model
BWD_COMPUTE_MULTIPLIER: float = 2.0  # backward ~ 2x forward bytes (reference :36)
WEIGHTED_KERNEL_MULTIPLIER: float = 1.1
DP_ELEMENTWISE_KERNELS_PERF_FACTOR: float = 9.22

POOLING_FACTOR: float = 1.0  # default ids/sample per feature
BATCH_SIZE: int = 8192

# fraction of HBM the planner may fill with shards (rest: activations, comms
# buffers, allocator slack)
MAX_HBM_UTILIZATION: float = 0.92
