"""Control-plane RPC: framed binary messages over TCP.

MI355X re-design of the reference's two-sided SEND path
(RdmaRpcMsg.scala:40-78): messages are self-framing — ``len:u32 | type:u32``
header followed by the body — and are segmented into at most
``recv_wr_size``-byte frames exactly like the reference segments messages to
fit its receive work requests (RdmaRpcMsg.scala:42-58). On MI355X there is
no receive-WR size limit (TCP streams), but the segmenting codec is kept:
it bounds per-message buffer use and is contract-tested.

Message inventory (reference has Hello + Announce, RdmaRpcMsg.scala:29-32;
the rebuild adds shuffle lifecycle because there is no Spark driver-side
broadcast to smuggle handles through):

* HELLO      executor -> driver   executor identity + metadata segment
* ANNOUNCE   driver  -> all       full membership list + app id
* REGISTER   app     -> driver    create shuffle (numMaps, numPartitions)
* HANDLE     driver  -> app       shuffle handle (driver table coords)
* UNREGISTER app     -> driver    drop shuffle
* BYE        executor-> driver    clean detach
* LOOKUP     executor-> driver    verify a shuffle id's (numMaps, numParts)
* TABLE_READ executor-> driver    driver-table bytes (cross-host hop 1)
* TABLE_WRITE executor-> driver   12-byte publish (cross-host publish)
"""

from __future__ import annotations

import socket
import struct
import threading
from dataclasses import dataclass
from typing import List, Optional

MSG_HELLO = 0
MSG_ANNOUNCE = 1
MSG_REGISTER = 2
MSG_HANDLE = 3
MSG_UNREGISTER = 4
MSG_BYE = 5
MSG_ERROR = 6
MSG_BARRIER = 7
MSG_BARRIER_OK = 8
MSG_LOOKUP = 9        # body: shuffle_id i32 -> HANDLE reply (or ERROR)
MSG_TABLE_READ = 10   # body: shuffle_id i32 -> TABLE_DATA reply
MSG_TABLE_DATA = 11   # body: shuffle_id i32 + raw table bytes
MSG_TABLE_WRITE = 12  # body: shuffle_id i32, map_id i32, addr u64, key u32

_HDR = struct.Struct("<III")  # frame_len (incl. header), msg_type, total_body_len


def _pack_str(s: str) -> bytes:
    b = s.encode("utf-8")
    if len(b) > 0xFFFF:
        raise ValueError("string too long")
    return struct.pack("<H", len(b)) + b


def _unpack_str(buf: bytes, off: int) -> tuple:
    (n,) = struct.unpack_from("<H", buf, off)
    off += 2
    return buf[off:off + n].decode("utf-8"), off + n


@dataclass(frozen=True)
class ExecutorInfo:
    """Identity of one executor — reference RdmaShuffleManagerId
    (RdmaUtils.scala:74-143): (host, port, blockManagerId) becomes
    (executor_id, host, port, gpu_id, meta_segment)."""
    executor_id: int
    host: str
    port: int
    gpu_id: int
    meta_segment: str   # uri of the executor's metadata segment

    def pack(self) -> bytes:
        return (struct.pack("<HHh", self.executor_id, self.port, self.gpu_id)
                + _pack_str(self.host) + _pack_str(self.meta_segment))

    @classmethod
    def unpack_from(cls, buf: bytes, off: int) -> tuple:
        executor_id, port, gpu_id = struct.unpack_from("<HHh", buf, off)
        off += 6
        host, off = _unpack_str(buf, off)
        meta, off = _unpack_str(buf, off)
        return cls(executor_id, host, port, gpu_id, meta), off


# ---------------------------------------------------------------------------
# message bodies


def pack_hello(info: ExecutorInfo) -> bytes:
    return info.pack()


def unpack_hello(body: bytes) -> ExecutorInfo:
    info, _ = ExecutorInfo.unpack_from(body, 0)
    return info


def pack_announce(app_id: str, members: List[ExecutorInfo]) -> bytes:
    out = [_pack_str(app_id), struct.pack("<I", len(members))]
    out += [m.pack() for m in members]
    return b"".join(out)


def unpack_announce(body: bytes) -> tuple:
    app_id, off = _unpack_str(body, 0)
    (n,) = struct.unpack_from("<I", body, off)
    off += 4
    members = []
    for _ in range(n):
        m, off = ExecutorInfo.unpack_from(body, off)
        members.append(m)
    return app_id, members


_REGISTER = struct.Struct("<iii")  # shuffle_id, num_maps, num_partitions


def pack_register(shuffle_id: int, num_maps: int, num_partitions: int) -> bytes:
    return _REGISTER.pack(shuffle_id, num_maps, num_partitions)


def unpack_register(body: bytes) -> tuple:
    return _REGISTER.unpack(body)


def pack_handle(shuffle_id: int, num_maps: int, num_partitions: int,
                table_uri: str, table_addr: int) -> bytes:
    """Shuffle handle: carries the driver table's coordinates to executors,
    the reference's RdmaBaseShuffleHandle trick (RdmaUtils.scala:145-151)."""
    return (struct.pack("<iiiQ", shuffle_id, num_maps, num_partitions, table_addr)
            + _pack_str(table_uri))


def unpack_handle(body: bytes) -> tuple:
    shuffle_id, num_maps, num_partitions, table_addr = struct.unpack_from("<iiiQ", body, 0)
    uri, _ = _unpack_str(body, 20)
    return shuffle_id, num_maps, num_partitions, uri, table_addr


def pack_unregister(shuffle_id: int) -> bytes:
    return struct.pack("<i", shuffle_id)


def unpack_unregister(body: bytes) -> int:
    return struct.unpack("<i", body)[0]


_TABLE_WRITE = struct.Struct("<iiQI")  # shuffle_id, map_id, addr, key


def pack_table_write(shuffle_id: int, map_id: int, addr: int, key: int) -> bytes:
    """Cross-host publish: the 12-byte (addr, key) driver-table entry
    carried over RPC instead of a local one-sided write — same layout as
    the reference's 12 B RDMA WRITE (RdmaShuffleManager.scala:410-412)."""
    return _TABLE_WRITE.pack(shuffle_id, map_id, addr, key)


def unpack_table_write(body: bytes) -> tuple:
    return _TABLE_WRITE.unpack(body)


def pack_table_data(shuffle_id: int, table: bytes) -> bytes:
    return struct.pack("<i", shuffle_id) + table


def unpack_table_data(body: bytes) -> tuple:
    (sid,) = struct.unpack_from("<i", body, 0)
    return sid, body[4:]


# ---------------------------------------------------------------------------
# framing / segmentation


def encode_frames(msg_type: int, body: bytes, max_frame: int) -> List[bytes]:
    """Split one message into self-framing segments of <= max_frame bytes.

    Mirrors the reference's segmentation to recvWrSize buffers
    (RdmaRpcMsg.scala:42-58): every segment carries (frame_len, msg_type,
    total_body_len); continuation segments set the high bit of msg_type.
    total_body_len lets the receiver preallocate the reassembly buffer.
    """
    if max_frame <= _HDR.size:
        raise ValueError("max_frame too small")
    payload_max = max_frame - _HDR.size
    frames = []
    first = True
    off = 0
    while True:
        chunk = body[off:off + payload_max]
        mtype = msg_type if first else (msg_type | 0x8000_0000)
        first = False
        frames.append(_HDR.pack(len(chunk) + _HDR.size, mtype, len(body)) + chunk)
        off += len(chunk)
        if off >= len(body):
            break
    return frames


class FrameDecoder:
    """Incremental decoder: feed bytes, yields (msg_type, body) messages."""

    def __init__(self):
        self._buf = bytearray()
        self._cur_type: Optional[int] = None
        self._cur_total = 0
        self._cur_body = bytearray()

    def feed(self, data: bytes) -> List[tuple]:
        self._buf.extend(data)
        out = []
        while len(self._buf) >= _HDR.size:
            frame_len, mtype, total = _HDR.unpack_from(self._buf, 0)
            if len(self._buf) < frame_len:
                break
            payload = bytes(self._buf[_HDR.size:frame_len])
            del self._buf[:frame_len]
            if mtype & 0x8000_0000:  # continuation
                self._cur_body.extend(payload)
            else:  # first (possibly only) segment
                self._cur_type = mtype
                self._cur_total = total
                self._cur_body = bytearray(payload)
            if self._cur_type is not None and len(self._cur_body) >= self._cur_total:
                out.append((self._cur_type, bytes(self._cur_body)))
                self._cur_type = None
                self._cur_body = bytearray()
        return out


# ---------------------------------------------------------------------------
# connection helpers


class MsgConnection:
    """Blocking framed-message connection (thread-safe send)."""

    def __init__(self, sock: socket.socket, max_frame: int):
        self.sock = sock
        self.max_frame = max_frame
        self._decoder = FrameDecoder()
        self._send_lock = threading.Lock()
        try:
            self.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        except OSError:
            pass

    def send(self, msg_type: int, body: bytes) -> None:
        frames = encode_frames(msg_type, body, self.max_frame)
        with self._send_lock:
            self.sock.sendall(b"".join(frames))

    def recv(self, timeout: Optional[float] = None):
        """Receive one complete message, or None on EOF. Messages decoded
        beyond the first in one TCP read are queued and returned in order
        by later calls."""
        pending = getattr(self, "_pending", None)
        if pending:
            return pending.pop(0)
        self.sock.settimeout(timeout)
        while True:
            data = self.sock.recv(65536)
            if not data:
                return None
            msgs = self._decoder.feed(data)
            if msgs:
                if len(msgs) > 1:
                    self._pending = getattr(self, "_pending", [])
                    self._pending.extend(msgs[1:])
                return msgs[0]

    # kept as an alias for call-site clarity (drains queued messages first)
    recv_any = recv

    def close(self) -> None:
        try:
            self.sock.shutdown(socket.SHUT_RDWR)
        except OSError:
            pass
        self.sock.close()
