"""TCP data server — the cross-HOST read path.

On one node every read is genuinely one-sided (shm pread / xGMI peer
copy). Across hosts there is no shared fabric in this environment, so
each executor runs a small data server that answers
``(key, addr, length)`` requests from its own segments — host blocks are
pread, HBM blocks staged D2H. This is the fallback lane; the reference's
equivalent role is played by the NIC doing RDMA READ across machines.

Protocol: request ``<IQQ`` (key, addr, length); response ``<q`` status
(=length served, or negative errno) followed by the payload.
"""

from __future__ import annotations

import logging
import socket
import struct
import threading
from typing import Dict

from .map_output import split_key

log = logging.getLogger(__name__)

_REQ = struct.Struct("<IQQ")
_RSP = struct.Struct("<q")
MAX_READ = 4 << 30


class DataServer:
    def __init__(self, manager, host: str = "0.0.0.0", port: int = 0):
        self.manager = manager
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._srv.bind((host, port))
        self._srv.listen(64)
        self.port = self._srv.getsockname()[1]
        self._stopped = threading.Event()
        threading.Thread(target=self._accept_loop,
                         name="sparkrdma-dataserver", daemon=True).start()

    def _accept_loop(self) -> None:
        while not self._stopped.is_set():
            try:
                sock, _ = self._srv.accept()
            except OSError:
                return
            threading.Thread(target=self._serve, args=(sock,),
                             name="sparkrdma-dataserver-conn",
                             daemon=True).start()

    def _serve(self, sock: socket.socket) -> None:
        try:
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            while not self._stopped.is_set():
                hdr = self._recv_exact(sock, _REQ.size)
                if hdr is None:
                    return
                key, addr, length = _REQ.unpack(hdr)
                try:
                    if length > MAX_READ:
                        raise ValueError("read too large")
                    data = self._read_local(key, addr, length)
                    sock.sendall(_RSP.pack(len(data)))
                    sock.sendall(data)
                except Exception as e:  # report, keep serving
                    log.warning("data server read failed: %s", e)
                    try:
                        sock.sendall(_RSP.pack(-1))
                    except OSError:
                        return
        except OSError:
            pass
        finally:
            sock.close()

    def _read_local(self, key: int, addr: int, length: int) -> bytes:
        """Serve a read of OUR memory (host segment or HBM slab)."""
        mgr = self.manager
        exec_id, seg_id = split_key(key)
        if exec_id != mgr.executor_id:
            raise ValueError(f"key {key:#x} not served by executor "
                             f"{mgr.executor_id}")
        if seg_id & 0x8000:  # HBM slab: stage D2H
            base = mgr.gpu.local_base(seg_id)
            buf = bytearray(length)
            import numpy as np
            arr = np.frombuffer(buf, dtype=np.uint8)
            mgr.gpu.hs.memcpy_d2h(arr.ctypes.data, base + addr, length)
            return bytes(buf)
        return mgr._registry.read(key, addr, length)

    @staticmethod
    def _recv_exact(sock, n):
        buf = b""
        while len(buf) < n:
            chunk = sock.recv(n - len(buf))
            if not chunk:
                return None
            buf += chunk
        return buf

    def stop(self) -> None:
        self._stopped.set()
        try:
            self._srv.close()
        except OSError:
            pass


class DataClient:
    """Pooled client connections to peers' data servers, one per peer."""

    def __init__(self):
        self._conns: Dict[tuple, socket.socket] = {}
        self._locks: Dict[tuple, threading.Lock] = {}
        self._lock = threading.Lock()

    def read(self, host: str, port: int, key: int, addr: int,
             length: int) -> bytes:
        ep = (host, port)
        with self._lock:
            lock = self._locks.setdefault(ep, threading.Lock())
        with lock:
            sock = self._conns.get(ep)
            if sock is None:
                sock = socket.create_connection(ep, timeout=30)
                sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                self._conns[ep] = sock
            try:
                sock.sendall(_REQ.pack(key, addr, length))
                hdr = DataServer._recv_exact(sock, _RSP.size)
                if hdr is None:
                    raise ConnectionError("data server closed connection")
                (status,) = _RSP.unpack(hdr)
                if status < 0:
                    raise IOError(f"remote read failed (status {status})")
                data = DataServer._recv_exact(sock, status)
                if data is None:
                    raise ConnectionError("short data from server")
                return data
            except (OSError, ConnectionError):
                # drop the pooled connection; caller's retry semantics are
                # the fetcher's (failure fails the task)
                self._conns.pop(ep, None)
                try:
                    sock.close()
                except OSError:
                    pass
                raise

    def close(self) -> None:
        with self._lock:
            for s in self._conns.values():
                try:
                    s.close()
                except OSError:
                    pass
            self._conns.clear()
