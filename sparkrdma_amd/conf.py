"""Typed, range-validated shuffle configuration.

MI355X-native re-design of the reference's ``RdmaShuffleConf``
(reference: RdmaShuffleConf.scala:36-142). The key namespace
``spark.shuffle.rdma.*`` is preserved so existing reference configs read
unchanged; MI355X-specific keys are added under the same prefix.
"""

from __future__ import annotations

import os
import re
from dataclasses import dataclass, field
from typing import Any, Dict

_PREFIX = "spark.shuffle.rdma."

_SIZE_RE = re.compile(r"^\s*(\d+(?:\.\d+)?)\s*([kmgt]?)b?\s*$", re.IGNORECASE)
_SIZE_MULT = {"": 1, "k": 1 << 10, "m": 1 << 20, "g": 1 << 30, "t": 1 << 40}


def parse_bytes(value: Any) -> int:
    """Parse a Spark-style byte-size string ('256k', '48m', '10g') to bytes."""
    if isinstance(value, int):
        return value
    m = _SIZE_RE.match(str(value))
    if not m:
        raise ValueError(f"invalid byte size: {value!r}")
    return int(float(m.group(1)) * _SIZE_MULT[m.group(2).lower()])


def format_bytes(n: int) -> str:
    for suffix, mult in (("g", 1 << 30), ("m", 1 << 20), ("k", 1 << 10)):
        if n >= mult and n % mult == 0:
            return f"{n // mult}{suffix}"
    return str(n)


class ConfError(ValueError):
    pass


@dataclass
class ShuffleConf:
    """All tunables of the MI355X shuffle engine.

    Reference key inventory: RdmaShuffleConf.scala:61-142. Keys that made
    sense only for ibverbs hardware (ODP, CM timeouts) keep their names but
    are either repurposed or accepted-and-ignored with a note, so reference
    configs remain valid.
    """

    # --- transport / flow control (reference :61-85) ---
    recv_queue_depth: int = 256
    send_queue_depth: int = 4096
    recv_wr_size: int = 4096               # RPC segment size (framing codec)
    sw_flow_control: bool = True
    max_buffer_allocation_size: int = 10 << 30   # pool trim threshold
    device_num: int = 0                     # GPU ordinal (was: IB device index)
    use_odp: bool = False                   # accepted, no-op on HBM

    # --- poller/engine placement (reference :89) ---
    cpu_list: str = ""

    # --- write path (reference :94-95) ---
    shuffle_write_block_size: int = 8 << 20

    # --- read path (reference :100-118) ---
    shuffle_read_block_size: int = 256 << 10
    max_bytes_in_flight: int = 48 << 20
    pre_allocate_buffers: Dict[int, int] = field(default_factory=dict)

    # --- stats (reference :121-130) ---
    collect_shuffle_reader_stats: bool = False
    partition_location_fetch_timeout_ms: int = 120_000
    fetch_time_bucket_size_ms: int = 300
    fetch_time_num_buckets: int = 5
    collect_odp_stats: bool = True   # reference default (no-op on HBM: no ODP)

    # --- control plane (reference :134-142) ---
    driver_host: str = "127.0.0.1"
    driver_port: int = 0
    executor_port: int = 0
    rdma_cm_event_timeout_ms: int = 20_000
    teardown_listen_timeout_ms: int = 50
    resolve_path_timeout_ms: int = 2_000
    max_connection_attempts: int = 5

    # --- MI355X-native additions ---
    hbm_pool_size: int = 0            # 0 = auto (free HBM minus reserve)
    hbm_slab_size: int = 1 << 30      # slab granularity for IPC export
    gpu_id: int = -1                  # -1 = from LOCAL_RANK / device_num
    use_rccl: bool = False            # stage-mode alltoallv instead of peer copies
    transport: str = "auto"           # auto | shm | ipc | rccl
    shm_dir: str = "/dev/shm"
    read_requests_limit: int = 0      # 0 = send_queue_depth // cores
    executor_cores: int = 1
    # cross-host TCP lane (the one place a codec pays — PARITY.md): the
    # intra-node xGMI/shm paths never compress
    tcp_compress: bool = False        # zlib-1 per chunk
    tcp_chunk_size: int = 4 << 20     # streaming chunk (pinned D2H unit)

    def __post_init__(self) -> None:
        self._validate_range("recv_queue_depth", self.recv_queue_depth, 16, 1 << 20)
        self._validate_range("send_queue_depth", self.send_queue_depth, 16, 1 << 24)
        self._validate_range("recv_wr_size", self.recv_wr_size, 256, 1 << 24)
        self._validate_range("shuffle_write_block_size", self.shuffle_write_block_size,
                             4 << 10, 1 << 34)
        self._validate_range("shuffle_read_block_size", self.shuffle_read_block_size,
                             4 << 10, 1 << 32)
        self._validate_range("max_bytes_in_flight", self.max_bytes_in_flight,
                             self.shuffle_read_block_size, 1 << 40)
        if self.transport not in ("auto", "shm", "ipc", "rccl", "tcp"):
            raise ConfError(
                f"transport must be auto|shm|ipc|rccl|tcp, got {self.transport!r}")

    @staticmethod
    def _validate_range(name: str, value: int, lo: int, hi: int) -> None:
        if not (lo <= value <= hi):
            raise ConfError(f"{_PREFIX}{name}: {value} outside [{lo}, {hi}]")

    # --- construction from a Spark-style flat dict ---

    _KEYMAP = {
        "recvQueueDepth": ("recv_queue_depth", int),
        "sendQueueDepth": ("send_queue_depth", int),
        "recvWrSize": ("recv_wr_size", parse_bytes),
        "swFlowControl": ("sw_flow_control", None),
        "maxBufferAllocationSize": ("max_buffer_allocation_size", parse_bytes),
        "device.num": ("device_num", int),
        "useOdp": ("use_odp", None),
        "cpuList": ("cpu_list", str),
        "shuffleWriteBlockSize": ("shuffle_write_block_size", parse_bytes),
        "shuffleReadBlockSize": ("shuffle_read_block_size", parse_bytes),
        "maxBytesInFlight": ("max_bytes_in_flight", parse_bytes),
        "collectShuffleReaderStats": ("collect_shuffle_reader_stats", None),
        "partitionLocationFetchTimeout": ("partition_location_fetch_timeout_ms", int),
        "fetchTimeBucketSizeInMs": ("fetch_time_bucket_size_ms", int),
        "fetchTimeNumBuckets": ("fetch_time_num_buckets", int),
        "collectOdpStats": ("collect_odp_stats", None),
        "driverHost": ("driver_host", str),
        "driverPort": ("driver_port", int),
        "executorPort": ("executor_port", int),
        "rdmaCmEventTimeout": ("rdma_cm_event_timeout_ms", int),
        "teardownListenTimeout": ("teardown_listen_timeout_ms", int),
        "resolvePathTimeout": ("resolve_path_timeout_ms", int),
        "maxConnectionAttempts": ("max_connection_attempts", int),
        "hbmPoolSize": ("hbm_pool_size", parse_bytes),
        "hbmSlabSize": ("hbm_slab_size", parse_bytes),
        "gpuId": ("gpu_id", int),
        "useRccl": ("use_rccl", None),
        "transport": ("transport", str),
        "shmDir": ("shm_dir", str),
        "rdmaReadRequestsLimit": ("read_requests_limit", int),
        "executorCores": ("executor_cores", int),
        "tcpCompress": ("tcp_compress", None),
        "tcpChunkSize": ("tcp_chunk_size", parse_bytes),
    }

    @classmethod
    def from_dict(cls, conf: Dict[str, Any]) -> "ShuffleConf":
        kwargs: Dict[str, Any] = {}
        for key, raw in conf.items():
            if not key.startswith(_PREFIX):
                continue
            short = key[len(_PREFIX):]
            if short == "preAllocateBuffers":
                kwargs["pre_allocate_buffers"] = _parse_prealloc(raw)
                continue
            if short not in cls._KEYMAP:
                raise ConfError(f"unknown key {key}")
            attr, conv = cls._KEYMAP[short]
            if conv is None:  # bool
                kwargs[attr] = str(raw).lower() in ("1", "true", "yes")
            else:
                kwargs[attr] = conv(raw)
        return cls(**kwargs)

    def resolved_gpu_id(self) -> int:
        if self.gpu_id >= 0:
            return self.gpu_id
        lr = os.environ.get("LOCAL_RANK")
        if lr is not None:
            return int(lr)
        return self.device_num

    def resolved_read_requests_limit(self) -> int:
        if self.read_requests_limit > 0:
            return self.read_requests_limit
        # reference: sendQueueDepth / executor cores (RdmaShuffleFetcherIterator.scala:82-83)
        return max(1, self.send_queue_depth // max(1, self.executor_cores))


def _parse_prealloc(spec: str) -> Dict[int, int]:
    """Parse 'size:count,size:count' (reference RdmaShuffleConf.scala:104-118)."""
    out: Dict[int, int] = {}
    if not spec:
        return out
    for part in str(spec).split(","):
        size_s, count_s = part.split(":")
        out[parse_bytes(size_s)] = int(count_s)
    return out
