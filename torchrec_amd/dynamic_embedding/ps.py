"""Dynamic-embedding parameter-server bridge with pluggable IO transports.

Reference parity: contrib/dynamic_embedding `tde/ps.cpp` + `details/
io_registry.cpp` (+ `redis_io.cpp`) — evicted rows are PUSHED to an external
store and re-admitted ids PULL their rows back, with the transport chosen by
name from a registry. This image has no Redis, so the in-repo transports are
``file`` (append-only row log, the SSD tier's backend) and ``memory``; a
production transport registers itself with :func:`register_ps_io`.
"""

from __future__ import annotations

import abc
import os
import tempfile
from typing import Callable, Dict, List, Optional

import numpy as np
import torch


class PSIO(abc.ABC):
    """One table's external row store: rows are (dim,) fp32 + scalar state."""

    @abc.abstractmethod
    def push(self, ids: List[int], rows: np.ndarray, state: np.ndarray) -> None:
        ...

    @abc.abstractmethod
    def pull(self, id_: int) -> Optional[tuple]:
        """-> (row, state) or None if the id was never pushed."""

    def close(self) -> None:  # pragma: no cover - transport hook
        pass


class MemoryPSIO(PSIO):
    def __init__(self, dim: int, table: int = 0) -> None:
        self._store: Dict[int, tuple] = {}

    def push(self, ids, rows, state) -> None:
        for k, i in enumerate(ids):
            self._store[int(i)] = (rows[k].copy(), float(state[k]))

    def pull(self, id_):
        return self._store.pop(int(id_), None)


class FilePSIO(PSIO):
    """Append-only on-disk row log + in-memory offset index (the SSD tier)."""

    def __init__(self, dim: int, path: Optional[str] = None, table: int = 0) -> None:
        self._dim = dim
        # a shared explicit path is suffixed per table so logs don't clobber
        self._path = (f"{path}.t{table}" if path else None) or os.path.join(
            tempfile.mkdtemp(prefix="trec_amd_ps_"), "rows.log"
        )
        os.makedirs(os.path.dirname(self._path), exist_ok=True)
        self._f = open(self._path, "w+b")
        self._index: Dict[int, int] = {}

    def push(self, ids, rows, state) -> None:
        for k, i in enumerate(ids):
            self._f.seek(0, 2)
            off = self._f.tell()
            self._f.write(rows[k].astype(np.float32).tobytes())
            self._f.write(np.float32(state[k]).tobytes())
            self._index[int(i)] = off

    def pull(self, id_):
        off = self._index.pop(int(id_), None)
        if off is None:
            return None
        self._f.seek(off)
        buf = self._f.read(self._dim * 4 + 4)
        row = np.frombuffer(buf[: self._dim * 4], dtype=np.float32).copy()
        st = float(np.frombuffer(buf[self._dim * 4 :], dtype=np.float32)[0])
        return row, st

    def close(self) -> None:
        self._f.close()


_IO_REGISTRY: Dict[str, Callable[..., PSIO]] = {
    "memory": MemoryPSIO,
    "file": FilePSIO,
}


def register_ps_io(name: str, factory: Callable[..., PSIO]) -> None:
    """Register a transport (reference io_registry.cpp register_io)."""
    _IO_REGISTRY[name] = factory


def get_ps_io(name: str, dim: int, **kwargs) -> PSIO:
    if name == "tcp" and name not in _IO_REGISTRY:
        import torchrec_amd.dynamic_embedding.ps_net  # noqa: F401  registers "tcp"
    return _IO_REGISTRY[name](dim, **kwargs)


class ParameterServer:
    """Per-table PS endpoints over a chosen transport (reference tde/ps.cpp:
    fetch on admission, evict on displacement)."""

    def __init__(self, dims: List[int], io: str = "memory", **io_kwargs) -> None:
        self._ios = [get_ps_io(io, d, table=i, **io_kwargs) for i, d in enumerate(dims)]

    def evict(
        self, table: int, ids: torch.Tensor, rows: torch.Tensor, state: torch.Tensor
    ) -> None:
        self._ios[table].push(
            [int(i) for i in ids],
            rows.detach().float().cpu().numpy(),
            state.detach().float().cpu().numpy(),
        )

    def fetch(self, table: int, id_: int):
        return self._ios[table].pull(id_)

    def close(self) -> None:
        for io in self._ios:
            io.close()
