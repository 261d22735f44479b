"""Observability: fetch-latency histograms + task metrics.

Re-design of RdmaShuffleReaderStats.scala:32-81 — per-remote-executor and
global histograms of remote-fetch latency with configurable bucket width /
count — and of the inline Spark task-metrics calls
(RdmaShuffleFetcherIterator.scala:60,104-106,353-361).
ODP stats (sysfs page-fault counters, :83-99) have no HBM analog and are
dropped; the hook shape is kept via `extra` counters.
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict


class FetchHistogram:
    """Latency histogram: fetch_time_num_buckets buckets of
    fetch_time_bucket_size_ms each, plus overflow (reference
    RdmaRemoteFetchHistogram, RdmaShuffleReaderStats.scala:32-53)."""

    def __init__(self, bucket_ms: int, num_buckets: int):
        self.bucket_ms = bucket_ms
        self.buckets = [0] * (num_buckets + 1)

    def add(self, latency_ms: float) -> None:
        idx = min(int(latency_ms // self.bucket_ms), len(self.buckets) - 1)
        self.buckets[idx] += 1

    def format(self) -> str:
        parts = []
        for i, c in enumerate(self.buckets[:-1]):
            parts.append(f"[{i * self.bucket_ms}-{(i + 1) * self.bucket_ms}ms: {c}]")
        parts.append(f"[>{(len(self.buckets) - 1) * self.bucket_ms}ms: {self.buckets[-1]}]")
        return " ".join(parts)


class ShuffleReaderStats:
    def __init__(self, conf):
        self._bucket_ms = conf.fetch_time_bucket_size_ms
        self._n = conf.fetch_time_num_buckets
        self._global = FetchHistogram(self._bucket_ms, self._n)
        self._per_remote: Dict[int, FetchHistogram] = {}
        self._lock = threading.Lock()

    def update(self, remote_executor_id: int, latency_ms: float) -> None:
        with self._lock:
            h = self._per_remote.get(remote_executor_id)
            if h is None:
                h = FetchHistogram(self._bucket_ms, self._n)
                self._per_remote[remote_executor_id] = h
            h.add(latency_ms)
            self._global.add(latency_ms)

    def print_histograms(self, log) -> None:
        with self._lock:
            for rid, h in sorted(self._per_remote.items()):
                log.info("fetch latency from executor %d: %s", rid, h.format())
            log.info("fetch latency global: %s", self._global.format())


@dataclass
class TaskMetrics:
    """Per-task read/write metrics, the Spark task-metrics analog."""
    remote_blocks_fetched: int = 0
    local_blocks_fetched: int = 0
    remote_bytes_read: int = 0
    local_bytes_read: int = 0
    fetch_wait_ns: int = 0
    records_read: int = 0
    bytes_written: int = 0
    records_written: int = 0
    write_ns: int = 0
    extra: Dict[str, float] = field(default_factory=dict)

    def merge(self, other: "TaskMetrics") -> None:
        """Accumulate another task's counters (the executor-lifetime
        rollup the reference feeds into Spark's task metrics)."""
        self.remote_blocks_fetched += other.remote_blocks_fetched
        self.local_blocks_fetched += other.local_blocks_fetched
        self.remote_bytes_read += other.remote_bytes_read
        self.local_bytes_read += other.local_bytes_read
        self.fetch_wait_ns += other.fetch_wait_ns
        self.records_read += other.records_read
        self.bytes_written += other.bytes_written
        self.records_written += other.records_written
        self.write_ns += other.write_ns
        for k, v in other.extra.items():
            self.extra[k] = self.extra.get(k, 0.0) + v

    def format(self) -> str:
        return (f"read: {self.remote_bytes_read >> 20} MiB remote / "
                f"{self.local_bytes_read >> 20} MiB local "
                f"({self.remote_blocks_fetched}+{self.local_blocks_fetched} "
                f"blocks, fetch-wait {self.fetch_wait_ns / 1e9:.2f}s) | "
                f"write: {self.bytes_written >> 20} MiB, "
                f"{self.records_written} records, "
                f"{self.write_ns / 1e9:.2f}s")
