"""TCP data server — the cross-HOST read path.

On one node every read is genuinely one-sided (shm pread / xGMI peer
copy). Across hosts there is no shared fabric in this environment, so
each executor runs a small data server that answers
``(key, addr, length)`` requests from its own segments. This is the
fallback lane; the reference's equivalent role is played by the NIC
doing RDMA READ across machines.

r02 protocol (VERDICT r01 item 8 — "streaming + optional codec"):
responses STREAM in chunks instead of one monolithic buffer. HBM blocks
stage device->host per chunk through two PINNED buffers, double-buffered
so chunk i+1's async D2H overlaps chunk i's socket send; host-segment
chunks are pread per chunk (page cache does the pipelining). Each chunk
is optionally deflate-compressed (zlib-1) — the one lane where a codec
pays (PARITY.md: xGMI/HBM paths never compress, the reference's
wrapStream compressed exactly this kind of inter-node hop).

Wire format:
  request :  <IQQ>  key, addr, length
  response:  <qI>   status (=length served, or <0), flags (1 = deflate)
             then per chunk: <II> raw_len, wire_len, followed by wire_len
             bytes (deflate of the chunk, or the raw chunk) until the
             raw_lens sum to status.
"""

from __future__ import annotations

import logging
import socket
import struct
import threading
import zlib
from typing import Dict

from .map_output import split_key

log = logging.getLogger(__name__)


def _native():
    """The hipshuffle extension's GIL-free socket helpers (None when the
    extension is unavailable — pure-Python loops then serve)."""
    global _NATIVE
    if _NATIVE is False:
        return None
    if _NATIVE is None:
        try:
            from .ops import load
            _NATIVE = load(build_if_missing=False)
        except Exception:
            _NATIVE = False
            return None
    return _NATIVE


_NATIVE = None

_REQ = struct.Struct("<IQQ")
_RSP = struct.Struct("<qI")
_CHUNK = struct.Struct("<II")
FLAG_DEFLATE = 1
MAX_READ = 4 << 30


class _PinnedPair:
    """Two pinned staging buffers for double-buffered D2H."""

    def __init__(self, hs, chunk: int):
        self.hs = hs
        self.chunk = chunk
        self.ptrs = [hs.host_alloc_pinned(chunk) for _ in range(2)]

    def free(self) -> None:
        for p in self.ptrs:
            try:
                self.hs.host_free_pinned(p)
            except Exception:
                pass
        self.ptrs = []


class DataServer:
    def __init__(self, manager, host: str = "0.0.0.0", port: int = 0):
        self.manager = manager
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._srv.bind((host, port))
        self._srv.listen(64)
        self.port = self._srv.getsockname()[1]
        self._stopped = threading.Event()
        self._chunk = manager.conf.tcp_chunk_size
        self._flags = FLAG_DEFLATE if manager.conf.tcp_compress else 0
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
        pinned = None
        try:
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            try:
                sock.setsockopt(socket.SOL_SOCKET, socket.SO_SNDBUF, 8 << 20)
            except OSError:
                pass
            while not self._stopped.is_set():
                hdr = self._recv_exact(sock, _REQ.size)
                if hdr is None:
                    return
                key, addr, length = _REQ.unpack(hdr)
                # validate BEFORE the OK header: pre-header failures get a
                # clean -1 status; failures mid-stream can only close the
                # connection (the client surfaces a fetch failure)
                try:
                    if length > MAX_READ:
                        raise ValueError("read too large")
                    exec_id, seg_id = split_key(key)
                    if exec_id != self.manager.executor_id:
                        raise ValueError(
                            f"key {key:#x} not served by executor "
                            f"{self.manager.executor_id}")
                except Exception as e:
                    log.warning("data server rejected read: %s", e)
                    try:
                        sock.sendall(_RSP.pack(-1, 0))
                        continue
                    except OSError:
                        return
                try:
                    sock.sendall(_RSP.pack(length, self._flags))
                    if seg_id & 0x8000:
                        pinned = self._stream_hbm(sock, seg_id, addr,
                                                  length, pinned)
                    else:
                        self._stream_host(sock, key, addr, length)
                except (BrokenPipeError, ConnectionResetError):
                    return
                except Exception as e:  # mid-stream: unrecoverable here
                    log.warning("data server stream failed: %s", e)
                    return
        except OSError:
            pass
        finally:
            if pinned is not None:
                pinned.free()
            sock.close()

    def _send_chunk(self, sock: socket.socket, raw: bytes) -> None:
        if self._flags & FLAG_DEFLATE:
            wire = zlib.compress(raw, 1)
            if len(wire) >= len(raw):   # incompressible: send raw
                sock.sendall(_CHUNK.pack(len(raw), len(raw)))
                sock.sendall(raw)
                return
            sock.sendall(_CHUNK.pack(len(raw), len(wire)))
            sock.sendall(wire)
        else:
            sock.sendall(_CHUNK.pack(len(raw), len(raw)))
            sock.sendall(raw)

    def _stream_host(self, sock, key: int, addr: int, length: int) -> None:
        reader = self.manager._registry.reader(key)
        if not (self._flags & FLAG_DEFLATE):
            # zero-copy lane: kernel-side sendfile from the shm segment,
            # no userspace byte handling at all
            import os
            off = 0
            while off < length:
                c = min(self._chunk, length - off)
                sock.sendall(_CHUNK.pack(c, c))
                sent = 0
                while sent < c:
                    sent += os.sendfile(sock.fileno(), reader.fd,
                                        addr + off + sent, c - sent)
                off += c
            return
        off = 0
        while off < length:
            c = min(self._chunk, length - off)
            raw = reader.read(addr + off, c)
            if len(raw) != c:
                raise IOError(f"short segment read {len(raw)}/{c}")
            self._send_chunk(sock, raw)
            off += c

    def _stream_hbm(self, sock, seg_id: int, addr: int, length: int,
                    pinned):
        """Device blocks: async D2H into pinned buffer B while buffer A's
        bytes are on the socket — the copy/send pipeline. Sends are
        ZERO-COPY views over the pinned buffers (safe: the next reuse of
        a buffer waits on its event only after the send returned)."""
        import ctypes
        gpu = self.manager.gpu
        hs = gpu.hs
        hs.set_device(gpu.device)
        base = gpu.local_base(seg_id)
        if pinned is None or pinned.chunk < self._chunk:
            if pinned is not None:
                pinned.free()
            pinned = _PinnedPair(hs, self._chunk)
        slot = 63   # dedicated staging stream slot in the copy engine
        offs = list(range(0, length, self._chunk))
        evs = [None, None]

        def send_slot(idx, nbytes):
            hs.wait_event(evs[idx])
            if not (self._flags & FLAG_DEFLATE):
                # GIL-free raw path: header + payload straight from the
                # pinned buffer
                sock.sendall(_CHUNK.pack(nbytes, nbytes))
                if hs.tcp_send_all(sock.fileno(), pinned.ptrs[idx],
                                   nbytes) < 0:
                    raise ConnectionError("send failed")
                return
            view = (ctypes.c_char * nbytes).from_address(pinned.ptrs[idx])
            self._send_chunk(sock, view)

        for i, off in enumerate(offs):
            c = min(self._chunk, length - off)
            evs[i % 2] = hs.read_batch(slot, [pinned.ptrs[i % 2]],
                                       [base + addr + off], [c])
            if i > 0:
                pc = min(self._chunk, length - offs[i - 1])
                send_slot((i - 1) % 2, pc)
        if offs:
            last = len(offs) - 1
            send_slot(last % 2, min(self._chunk, length - offs[last]))
        return pinned

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
    """Client connections to peers' data servers — a small POOL per peer
    so concurrent fetch workers fan in over parallel sockets instead of
    serializing on one."""

    MAX_CONNS_PER_PEER = 4

    def __init__(self):
        self._free: Dict[tuple, list] = {}
        self._counts: Dict[tuple, int] = {}
        self._lock = threading.Lock()
        self._cv = threading.Condition(self._lock)

    def _acquire(self, ep) -> socket.socket:
        with self._cv:
            while True:
                free = self._free.setdefault(ep, [])
                if free:
                    return free.pop()
                if self._counts.get(ep, 0) < self.MAX_CONNS_PER_PEER:
                    self._counts[ep] = self._counts.get(ep, 0) + 1
                    break
                self._cv.wait(1.0)
        try:
            sock = socket.create_connection(ep, timeout=30)
            # BLOCKING mode with a kernel-level receive timeout: python
            # timeout-mode sockets are O_NONBLOCK underneath, which the
            # GIL-free native recv loop cannot use; SO_RCVTIMEO keeps the
            # hang protection for both the native and python paths
            sock.settimeout(None)
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            try:
                sock.setsockopt(socket.SOL_SOCKET, socket.SO_RCVTIMEO,
                                struct.pack("ll", 30, 0))
                sock.setsockopt(socket.SOL_SOCKET, socket.SO_RCVBUF, 8 << 20)
            except OSError:
                pass
            return sock
        except BaseException:
            with self._cv:
                self._counts[ep] -= 1
                self._cv.notify()
            raise

    def _release(self, ep, sock, broken: bool) -> None:
        with self._cv:
            if broken:
                self._counts[ep] -= 1
                try:
                    sock.close()
                except OSError:
                    pass
            else:
                self._free.setdefault(ep, []).append(sock)
            self._cv.notify()

    def read(self, host: str, port: int, key: int, addr: int,
             length: int) -> bytes:
        ep = (host, port)
        sock = self._acquire(ep)
        released = False
        try:
            sock.sendall(_REQ.pack(key, addr, length))
            hdr = DataServer._recv_exact(sock, _RSP.size)
            if hdr is None:
                raise ConnectionError("data server closed connection")
            status, flags = _RSP.unpack(hdr)
            if status < 0:
                # protocol-level rejection: the connection itself is fine
                self._release(ep, sock, broken=False)
                released = True
                raise IOError(f"remote read failed (status {status})")
            out = bytearray(status)
            nat = _native()
            if status and not (flags & FLAG_DEFLATE) and nat is not None:
                # whole chunked body received by the GIL-free C loop
                import numpy as np
                addr = np.frombuffer(out, dtype=np.uint8).ctypes.data
                got = nat.tcp_recv_chunks(sock.fileno(), addr, status)
                if got != status:
                    raise ConnectionError(
                        f"native chunk receive failed ({got}/{status})")
                self._release(ep, sock, broken=False)
                released = True
                return bytes(out)
            mv = memoryview(out)
            off = 0
            while off < status:
                chdr = DataServer._recv_exact(sock, _CHUNK.size)
                if chdr is None:
                    raise ConnectionError("short chunk header")
                raw_len, wire_len = _CHUNK.unpack(chdr)
                if wire_len == raw_len:
                    # raw chunk: receive STRAIGHT into place (no copy)
                    self._recv_into(sock, mv[off:off + raw_len])
                else:
                    wire = DataServer._recv_exact(sock, wire_len)
                    if wire is None:
                        raise ConnectionError("short chunk payload")
                    dec = zlib.decompress(wire)
                    if len(dec) != raw_len:
                        raise IOError("chunk decompress length mismatch")
                    mv[off:off + raw_len] = dec
                off += raw_len
            self._release(ep, sock, broken=False)
            released = True
            return bytes(out)
        except BaseException:
            # ANY failure after acquire (socket error, zlib error, ...)
            # must return the pool slot exactly once — a leak here would
            # eventually deadlock _acquire at MAX_CONNS_PER_PEER
            if not released:
                self._release(ep, sock, broken=True)
            raise

    @staticmethod
    def _recv_into(sock, view) -> None:
        got = 0
        n = len(view)
        while got < n:
            r = sock.recv_into(view[got:])
            if r == 0:
                raise ConnectionError("short chunk payload")
            got += r

    def close(self) -> None:
        with self._cv:
            for conns in self._free.values():
                for s in conns:
                    try:
                        s.close()
                    except OSError:
                        pass
            self._free.clear()
            self._counts.clear()
