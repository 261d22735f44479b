"""Workload correctness on CPU (single-process engine; multi-process
coverage in test_multiprocess.py)."""

import numpy as np
import pytest

from sparkrdma_amd.conf import ShuffleConf
from sparkrdma_amd.engine import Engine


@pytest.fixture
def engine(tmp_path):
    conf = ShuffleConf(shm_dir=str(tmp_path), max_buffer_allocation_size=1 << 30)
    eng = Engine(conf, rank=0, world_size=1, driver_port=0)
    yield eng
    eng.shutdown()


def test_terasort_cpu(engine):
    from sparkrdma_amd.workloads.terasort import TeraSort
    ts = TeraSort(engine, records_per_executor=50_000,
                  partitions_per_executor=16, device="cpu", validate=True)
    r = ts.run_step()
    assert r.records == 50_000
    r2 = ts.run_step()  # repeatable
    assert r2.records == 50_000


def test_pagerank_cpu_matches_dense_reference(engine):
    from sparkrdma_amd.workloads.pagerank import PageRank
    V, E, iters = 1 << 10, 20_000, 4
    pr = PageRank(engine, num_vertices=V, edges_per_executor=E,
                  partitions_per_executor=16, device="cpu",
                  iterations=iters, seed=3)
    res = pr.run_step()
    assert res.iterations == iters
    # rebuild the same edge list for the oracle
    rng = np.random.default_rng(3 * 7919 + 0)
    src = rng.integers(0, V, E, dtype=np.uint64).astype(np.int64)
    dst = rng.integers(0, V, E, dtype=np.uint64).astype(np.int64)
    want = PageRank.dense_reference(V, src, dst, iters)
    np.testing.assert_allclose(pr.ranks, want, rtol=1e-12)
    assert abs(pr.ranks.sum()) > 0


def test_sql_join_cpu_oracle(engine):
    from sparkrdma_amd.workloads.sql_join import SortMergeJoin
    j = SortMergeJoin(engine, rows_per_executor=30_000,
                      partitions_per_executor=16, device="cpu",
                      key_space_bits=16,  # dense keyspace -> many matches
                      validate=True)
    r = j.run_step()
    assert r.matches > 0


def test_groupby_cpu(engine):
    from sparkrdma_amd.workloads.groupby import GroupByKey
    g = GroupByKey(engine, rows_per_executor=20_000, num_keys=97)
    r = g.run_step()
    assert r.groups == 97
    assert r.rows == 20_000


def test_reduce_by_key_cpu(engine):
    from sparkrdma_amd.workloads.reduce_by_key import ReduceByKey
    r = ReduceByKey(engine, rows_per_executor=30_000, num_keys=500,
                    partitions_per_executor=16, device="cpu", validate=True)
    res = r.run_step()
    assert res.groups == 500


def test_api_tour_example_runs():
    """examples/api_tour.py is living documentation — keep it green."""
    import subprocess
    import sys
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run([sys.executable, os.path.join(repo, "examples", "api_tour.py")],
                       capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "range-partitioned" in r.stdout


def test_soak_scripts_smoke():
    """The endurance soak scripts stay runnable (a few seconds each)."""
    import subprocess
    import sys
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "scripts", "soak_control_plane.py"),
         "0.05"], capture_output=True, text=True, timeout=120)
    assert r.returncode == 0 and "soak ok" in r.stdout, r.stderr[-1500:]
