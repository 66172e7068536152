"""Network transport for the parameter-server bridge.

Reference parity: contrib/dynamic_embedding `redis_io.cpp` registers a
network-backed IO with the transport registry so evicted rows live in an
external store shared by trainers. This image has no Redis server, so the
MI355X framework ships its own: :class:`PSNetServer` is a small threaded
TCP key-value service speaking a length-prefixed binary protocol, and
:class:`TcpPSIO` is the client-side transport (registered as ``"tcp"``).

Protocol (all little-endian):
  PUSH: b"PUSH" u32 table  u32 n  u32 dim  n*i64 ids  n*dim*f32 rows  n*f32 state
        -> b"\\x01"
  PULL: b"PULL" u32 table  i64 id  u32 dim
        -> b"\\x01" dim*f32 row  f32 state   (pop semantics)  |  b"\\x00"
"""

from __future__ import annotations

import socket
import socketserver
import struct
import threading
from typing import Dict, List, Optional, Tuple

import numpy as np

from torchrec_amd.dynamic_embedding.ps import PSIO, register_ps_io


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = b""
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("peer closed mid-message")
        buf += chunk
    return buf


class _Handler(socketserver.BaseRequestHandler):
    def handle(self) -> None:
        store: Dict[Tuple[int, int], bytes] = self.server.store  # type: ignore[attr-defined]
        lock: threading.Lock = self.server.store_lock  # type: ignore[attr-defined]
        sock = self.request
        try:
            while True:
                op = _recv_exact(sock, 4)
                if op == b"PUSH":
                    table, n, dim = struct.unpack("<III", _recv_exact(sock, 12))
                    ids = np.frombuffer(_recv_exact(sock, n * 8), dtype="<i8")
                    rows = _recv_exact(sock, n * dim * 4)
                    state = _recv_exact(sock, n * 4)
                    with lock:
                        for k in range(n):
                            store[(table, int(ids[k]))] = (
                                rows[k * dim * 4 : (k + 1) * dim * 4]
                                + state[k * 4 : (k + 1) * 4]
                            )
                    sock.sendall(b"\x01")
                elif op == b"PULL":
                    table, id_, dim = struct.unpack("<IqI", _recv_exact(sock, 16))
                    with lock:
                        blob = store.pop((table, id_), None)
                    if blob is None or len(blob) != dim * 4 + 4:
                        sock.sendall(b"\x00")
                    else:
                        sock.sendall(b"\x01" + blob)
                elif op == b"QUIT":
                    return
                else:
                    raise ValueError(f"unknown op {op!r}")
        except (ConnectionError, OSError):
            return


class PSNetServer:
    """In-process threaded TCP row store. ``address`` is usable once started."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0) -> None:
        self._srv = socketserver.ThreadingTCPServer(
            (host, port), _Handler, bind_and_activate=True
        )
        self._srv.daemon_threads = True
        self._srv.store = {}  # type: ignore[attr-defined]
        self._srv.store_lock = threading.Lock()  # type: ignore[attr-defined]
        self._thread = threading.Thread(target=self._srv.serve_forever, daemon=True)
        self._thread.start()

    @property
    def address(self) -> Tuple[str, int]:
        return self._srv.server_address[:2]

    def __len__(self) -> int:
        with self._srv.store_lock:  # type: ignore[attr-defined]
            return len(self._srv.store)  # type: ignore[attr-defined]

    def close(self) -> None:
        self._srv.shutdown()
        self._srv.server_close()
        self._thread.join(timeout=5)


class TcpPSIO(PSIO):
    """Client transport: one socket per table, keyed by an integer namespace."""

    def __init__(self, dim: int, address: Tuple[str, int], table: int = 0) -> None:
        self._dim = dim
        self._table = table
        self._sock = socket.create_connection(tuple(address), timeout=30)
        self._sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)

    def push(self, ids: List[int], rows: np.ndarray, state: np.ndarray) -> None:
        n = len(ids)
        if n == 0:
            return
        msg = (
            b"PUSH"
            + struct.pack("<III", self._table, n, self._dim)
            + np.asarray(ids, dtype="<i8").tobytes()
            + np.ascontiguousarray(rows, dtype="<f4").tobytes()
            + np.ascontiguousarray(state, dtype="<f4").tobytes()
        )
        self._sock.sendall(msg)
        if _recv_exact(self._sock, 1) != b"\x01":
            raise IOError("PS push rejected")

    def pull(self, id_: int) -> Optional[tuple]:
        self._sock.sendall(
            b"PULL" + struct.pack("<IqI", self._table, int(id_), self._dim)
        )
        found = _recv_exact(self._sock, 1)
        if found == b"\x00":
            return None
        blob = _recv_exact(self._sock, self._dim * 4 + 4)
        row = np.frombuffer(blob[: self._dim * 4], dtype="<f4").copy()
        st = float(np.frombuffer(blob[self._dim * 4 :], dtype="<f4")[0])
        return row, st

    def close(self) -> None:
        try:
            self._sock.sendall(b"QUIT")
        except OSError:
            pass
        self._sock.close()


register_ps_io("tcp", TcpPSIO)
