"""sparkrdma_amd — an MI355X-native one-sided shuffle framework.

A from-scratch rebuild of the capabilities of Mellanox/SparkRDMA v3.1
(reference: /root/reference) for AMD Instinct MI355X nodes: the ibverbs
one-sided RDMA data plane becomes host-shm + ROCm-IPC/xGMI peer copies
(optionally RCCL alltoallv), the CPU sort-shuffle writer becomes a CDNA4
radix-partition kernel writing straight into an HBM block pool, and the
driver remains a pure metadata registry.

Layering:
  manager/writer/reader  — plugin API surface (L1-L3)
  rpc/driver/map_output  — control plane + metadata (L4)
  segments/block_pool/ops— one-sided data plane + HIP kernels (L5)
  conf/stats             — config (L6) + observability (L7)
  engine/workloads       — SPMD runtime + benchmark workloads (no reference
                           analog: Spark itself played this role)
"""

__version__ = "0.2.0"

from .conf import ShuffleConf
from .engine import Engine
from .manager import ShuffleHandle, ShuffleManager
from .partitioner import HashPartitioner, RangePartitioner
from .reader import FetchFailedError, ShuffleReader
from .writer import ShuffleWriter

__all__ = [
    "ShuffleConf", "ShuffleManager", "ShuffleHandle", "Engine",
    "ShuffleReader", "ShuffleWriter", "FetchFailedError",
    "HashPartitioner", "RangePartitioner",
]
