"""Unit coverage for stats.py: histogram bucketing and metric rollups
(reference RdmaShuffleReaderStats.scala:32-53 bucket semantics)."""

from sparkrdma_amd.stats import FetchHistogram, ShuffleReaderStats, TaskMetrics


def test_histogram_bucket_edges():
    h = FetchHistogram(bucket_ms=300, num_buckets=5)
    h.add(0)        # first bucket
    h.add(299.9)    # still first
    h.add(300)      # second bucket (floor semantics)
    h.add(1499.9)   # last regular bucket
    h.add(1500)     # overflow
    h.add(10_000)   # overflow
    assert h.buckets == [2, 1, 0, 0, 1, 2]
    s = h.format()
    assert "[0-300ms: 2]" in s and "[>1500ms: 2]" in s


def test_reader_stats_per_remote():
    class C:
        fetch_time_bucket_size_ms = 100
        fetch_time_num_buckets = 3
    st = ShuffleReaderStats(C())
    st.update(1, 50)
    st.update(1, 250)
    st.update(2, 50)
    assert st._per_remote[1].buckets == [1, 0, 1, 0]
    assert st._per_remote[2].buckets == [1, 0, 0, 0]
    assert st._global.buckets == [2, 0, 1, 0]


def test_task_metrics_merge_including_extra():
    a = TaskMetrics(remote_bytes_read=10, records_written=3)
    a.extra["gpu_ms"] = 1.5
    b = TaskMetrics(remote_bytes_read=5, records_written=2)
    b.extra["gpu_ms"] = 2.5
    b.extra["spill"] = 1
    a.merge(b)
    assert a.remote_bytes_read == 15
    assert a.records_written == 5
    assert a.extra == {"gpu_ms": 4.0, "spill": 1}
