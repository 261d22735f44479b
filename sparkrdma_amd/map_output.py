"""Map-output location tables — the shuffle's one-sided metadata.

Wire layouts are kept bit-compatible with the reference
(RdmaMapTaskOutput.scala:25-27):

* per-map-task table: one 16-byte entry per reduce partition
  ``addr:u64 | len:u32 | key:u32``  (ENTRY_SIZE = 16)
* driver-side table: one 12-byte entry per map task
  ``addr:u64 | key:u32``            (MAP_ENTRY_SIZE = 12)

On MI355X the (addr, key) pair no longer names an ibverbs memory region:
``key`` encodes ``(owner_executor_id << 16) | segment_id`` and ``addr`` is a
byte offset inside that segment — a segment being either a host
shared-memory file or an HBM slab exported once via ``hipIpcGetMemHandle``.
The fetcher resolves the key to an opened segment and performs a genuinely
one-sided read (shm pread / xGMI peer copy) with zero involvement of the
serving executor, preserving the reference's RDMA-READ property
(RdmaShuffleFetcherIterator.scala:174-177).
"""

from __future__ import annotations

import struct
from dataclasses import dataclass

import numpy as np

ENTRY_SIZE = 16       # per-partition entry in a map task's table
MAP_ENTRY_SIZE = 12   # per-map-task entry in the driver table

_ENTRY = struct.Struct("<QiI")      # addr, len, key
_MAP_ENTRY = struct.Struct("<QI")   # addr, key

ENTRY_DTYPE = np.dtype([("addr", "<u8"), ("len", "<i4"), ("key", "<u4")])
assert ENTRY_DTYPE.itemsize == ENTRY_SIZE


def make_key(executor_id: int, segment_id: int) -> int:
    if not (0 <= executor_id < (1 << 16)) or not (0 <= segment_id < (1 << 16)):
        raise ValueError(f"key fields out of range: exec={executor_id} seg={segment_id}")
    return (executor_id << 16) | segment_id


def split_key(key: int) -> tuple:
    return key >> 16, key & 0xFFFF


@dataclass(frozen=True)
class BlockLocation:
    """(address, length, key) of one shuffle block — reference RdmaUtils.scala:29-31."""
    addr: int
    length: int
    key: int

    def pack(self) -> bytes:
        return _ENTRY.pack(self.addr, self.length, self.key)

    @classmethod
    def unpack(cls, buf: bytes, offset: int = 0) -> "BlockLocation":
        addr, length, key = _ENTRY.unpack_from(buf, offset)
        return cls(addr, length, key)


class MapTaskOutput:
    """Per-map-task partition table: numPartitions × 16-byte entries.

    Backed by a numpy array over (possibly shared) memory so a remote
    fetcher can read the raw bytes directly (reference
    RdmaMapTaskOutput.scala:41-83).
    """

    def __init__(self, num_partitions: int, buffer: memoryview = None):
        self.num_partitions = num_partitions
        if buffer is None:
            self._arr = np.zeros(num_partitions, dtype=ENTRY_DTYPE)
        else:
            if len(buffer) < num_partitions * ENTRY_SIZE:
                raise ValueError("backing buffer too small")
            self._arr = np.frombuffer(
                buffer, dtype=ENTRY_DTYPE, count=num_partitions)

    def put(self, partition: int, addr: int, length: int, key: int) -> None:
        self._arr[partition] = (addr, length, key)

    def put_many(self, addrs, lengths, keys) -> None:
        self._arr["addr"] = addrs
        self._arr["len"] = lengths
        self._arr["key"] = keys

    def get(self, partition: int) -> BlockLocation:
        e = self._arr[partition]
        return BlockLocation(int(e["addr"]), int(e["len"]), int(e["key"]))

    def get_range(self, start: int, end: int) -> bytes:
        """Raw bytes of entries [start, end] inclusive — what a remote
        executor reads one-sidedly (reference RdmaMapTaskOutput.scala:66-83)."""
        return self._arr[start:end + 1].tobytes()

    def tobytes(self) -> bytes:
        return self._arr.tobytes()

    @property
    def nbytes(self) -> int:
        return self.num_partitions * ENTRY_SIZE

    @classmethod
    def from_bytes(cls, buf: bytes) -> "MapTaskOutput":
        n = len(buf) // ENTRY_SIZE
        out = cls(n)
        out._arr[:] = np.frombuffer(buf, dtype=ENTRY_DTYPE, count=n)
        return out

    @staticmethod
    def parse_locations(buf: bytes) -> list:
        """Parse a one-sided read of consecutive entries into BlockLocations."""
        arr = np.frombuffer(buf, dtype=ENTRY_DTYPE)
        return [BlockLocation(int(a), int(l), int(k))
                for a, l, k in zip(arr["addr"], arr["len"], arr["key"])]


class DriverTable:
    """Driver-held table: one 12-byte (addr, key) entry per map task, pointing
    at that map task's MapTaskOutput table (reference
    RdmaShuffleManager.scala:168-172, RdmaMapTaskOutput.scala:27).

    ``addr``/``key`` address the *table*, not data: key resolves to the
    owning executor's metadata segment, addr is the table's offset there.
    """

    def __init__(self, num_maps: int):
        self.num_maps = num_maps
        self._buf = bytearray(num_maps * MAP_ENTRY_SIZE)
        self._published = 0

    def publish(self, map_id: int, addr: int, key: int) -> None:
        if not (0 <= map_id < self.num_maps):
            raise IndexError(f"map_id {map_id} out of range")
        _MAP_ENTRY.pack_into(self._buf, map_id * MAP_ENTRY_SIZE, addr, key)
        self._published += 1

    def entry(self, map_id: int) -> tuple:
        return _MAP_ENTRY.unpack_from(self._buf, map_id * MAP_ENTRY_SIZE)

    def tobytes(self) -> bytes:
        return bytes(self._buf)

    @property
    def complete(self) -> bool:
        return self._published >= self.num_maps

    @staticmethod
    def parse(buf: bytes) -> list:
        """Parse the whole driver table into [(addr, key)] per map id."""
        n = len(buf) // MAP_ENTRY_SIZE
        return [_MAP_ENTRY.unpack_from(buf, i * MAP_ENTRY_SIZE) for i in range(n)]
