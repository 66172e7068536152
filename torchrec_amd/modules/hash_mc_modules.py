"""MPZCH: multi-probe zero-collision hashing on the GPU.

Reference parity: torchrec/modules/hash_mc_modules.py
(HashZchManagedCollisionModule :196, fbgemm zero_collision_hash :451).

Raw ids map to slots of a bounded identity table by multi-probe hashing: a
free slot is claimed atomically by the first id that probes it, so each
live id owns exactly one slot (zero collisions while capacity lasts); ids
that exhaust the probe budget share a fallback bucket. Eviction frees slots
whose last-seen stamp is stale. The device path is one HIP kernel
(cache.hip: hash_zch_kernel); the CPU path is a deterministic mirror."""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn

from torchrec_amd import ops
from torchrec_amd.sparse.jagged_tensor import JaggedTensor


def _splitmix64(x: torch.Tensor) -> torch.Tensor:
    mask = (1 << 64) - 1
    x = (x.to(torch.int64) + 0x9E3779B97F4A7C15) & mask
    # python-int domain to avoid int64 overflow semantics
    out = []
    for v in x.tolist():
        v &= mask
        v = ((v ^ (v >> 30)) * 0xBF58476D1CE4E5B9) & mask
        v = ((v ^ (v >> 27)) * 0x94D049BB133111EB) & mask
        out.append((v ^ (v >> 31)) & mask)
    return torch.tensor(out, dtype=torch.float64)  # exact up to 2^53 after mod


class HashZchManagedCollisionModule(nn.Module):
    def __init__(
        self,
        zch_size: int,
        device: Optional[torch.device] = None,
        max_probe: int = 128,
        eviction_interval: int = 1000,
    ) -> None:
        super().__init__()
        device = device or torch.device("cpu")
        self._zch_size = zch_size
        self._max_probe = min(max_probe, zch_size)
        self._eviction_interval = eviction_interval
        self._step = 0
        self.register_buffer(
            "identity", torch.full((zch_size,), -1, dtype=torch.int64, device=device)
        )
        self.register_buffer(
            "metadata", torch.zeros(zch_size, dtype=torch.int32, device=device)
        )

    def _remap_cpu(self, ids: torch.Tensor, train: bool) -> torch.Tensor:
        Z = self._zch_size
        out = torch.empty_like(ids)
        ident = self.identity
        hashes = _splitmix64(ids)
        for i in range(ids.numel()):
            iid = int(ids[i])
            h = int(hashes[i]) % Z
            slot = -1
            for p in range(self._max_probe):
                z = (h + p) % Z
                cur = int(ident[z])
                if cur == iid:
                    slot = z
                    break
                if cur == -1 and train:
                    ident[z] = iid
                    slot = z
                    break
            if slot < 0:
                slot = h
            if train:
                self.metadata[slot] = self._step
            out[i] = slot
        return out

    def remap(self, ids: torch.Tensor) -> torch.Tensor:
        self._step += 1
        if ids.is_cuda:
            ops.hip_ops()
            return torch.ops.trec_amd.hash_zch_remap(
                ids, self.identity, self.metadata, self._max_probe, self._step,
                self.training,
            )
        return self._remap_cpu(ids, self.training)

    def forward(self, features: JaggedTensor) -> JaggedTensor:
        return JaggedTensor(
            values=self.remap(features.values()),
            lengths=features.lengths(),
            weights=features.weights_or_none(),
        )

    def evict(self) -> Optional[torch.Tensor]:
        """Free slots unseen for an eviction interval; returns freed slots."""
        if self._step == 0 or self._step % self._eviction_interval != 0:
            return None
        # stale = not seen within the last `eviction_interval` steps
        older = self._step - self._eviction_interval + 1
        if self.identity.is_cuda:
            ops.hip_ops()
            ev = torch.ops.trec_amd.hash_zch_evict(self.identity, self.metadata, older)
            return ev[ev >= 0]
        mask = (self.identity != -1) & (self.metadata < older)
        slots = torch.nonzero(mask, as_tuple=True)[0]
        self.identity[slots] = -1
        self.metadata[slots] = 0
        return slots
