#!/usr/bin/env python3
"""Loopback bandwidth of the cross-host TCP lane (VERDICT r01 item 8
done-criterion: 'cross-host lane >= several GB/s on loopback, mixed
HBM/host blocks'). Serves one HBM block and one host block through the
data server on 127.0.0.1 and times repeated reads."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from sparkrdma_amd.conf import ShuffleConf
from sparkrdma_amd.driver import Driver
from sparkrdma_amd.manager import ShuffleManager
from sparkrdma_amd.map_output import make_key


def run(chunk_mb=4, size_mb=512, reps=5, compress=False):
    conf = ShuffleConf(transport="tcp", hbm_pool_size=4 << 30,
                       tcp_chunk_size=chunk_mb << 20,
                       tcp_compress=compress, gpu_id=0)
    driver = Driver(conf)
    m0 = ShuffleManager(conf, executor_id=0, driver_port=driver.port)
    m1 = ShuffleManager(conf, executor_id=1, driver_port=driver.port)
    try:
        size = size_mb << 20
        use_gpu = torch.cuda.is_available()
        results = {}
        # host block
        hblk = m1.pool.get(size)
        seg = m1.data_segment(hblk.segment_id)
        seg.write(0 + hblk.offset, b"\xab" * size)
        key = make_key(1, hblk.segment_id)
        m0._data_client.read("127.0.0.1", m1._data_server.port, key,
                             hblk.offset, size)  # warm
        t0 = time.perf_counter()
        for _ in range(reps):
            d = m0._data_client.read("127.0.0.1", m1._data_server.port,
                                     key, hblk.offset, size)
            assert len(d) == size
        dt = time.perf_counter() - t0
        results["host_gb_per_s"] = size * reps / dt / 1e9
        if use_gpu:
            gblk = m1.gpu.pool.get(size)
            base = m1.gpu.local_base(gblk.segment_id)
            t = torch.randint(-128, 127, (size,), dtype=torch.int8,
                              device="cuda")
            m1.gpu.hs.read_batch(0, [base + gblk.offset], [t.data_ptr()],
                                 [size])
            torch.cuda.synchronize()
            gkey = make_key(1, gblk.segment_id)
            m0._data_client.read("127.0.0.1", m1._data_server.port, gkey,
                                 gblk.offset, size)  # warm
            t0 = time.perf_counter()
            for _ in range(reps):
                d = m0._data_client.read("127.0.0.1", m1._data_server.port,
                                         gkey, gblk.offset, size)
                assert len(d) == size
            dt = time.perf_counter() - t0
            results["hbm_gb_per_s"] = size * reps / dt / 1e9
        print({"chunk_mb": chunk_mb, "size_mb": size_mb,
               "compress": compress,
               **{k: round(v, 2) for k, v in results.items()}})
    finally:
        m0.stop()
        m1.stop()
        driver.stop()


def run_parallel(nthreads=4, chunk_mb=1, size_mb=256, reps=4):
    """Aggregate lane bandwidth: the fetcher fans concurrent reads over
    the client's pooled connections (DataClient.MAX_CONNS_PER_PEER)."""
    import threading
    conf = ShuffleConf(transport="tcp", tcp_chunk_size=chunk_mb << 20)
    driver = Driver(conf)
    m0 = ShuffleManager(conf, executor_id=0, driver_port=driver.port)
    m1 = ShuffleManager(conf, executor_id=1, driver_port=driver.port)
    try:
        size = size_mb << 20
        blks = [m1.pool.get(size) for _ in range(nthreads)]
        keys = []
        for b in blks:
            m1.data_segment(b.segment_id).write(b.offset, b"\xcd" * size)
            keys.append((make_key(1, b.segment_id), b.offset))
        errs = []

        def worker(key, off):
            try:
                for _ in range(reps):
                    d = m0._data_client.read("127.0.0.1",
                                             m1._data_server.port,
                                             key, off, size)
                    assert len(d) == size
            except Exception as e:   # pragma: no cover
                errs.append(e)

        # warm connections
        for key, off in keys:
            m0._data_client.read("127.0.0.1", m1._data_server.port, key,
                                 off, 1 << 20)
        t0 = time.perf_counter()
        ts = [threading.Thread(target=worker, args=k)
              for k in keys]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        dt = time.perf_counter() - t0
        assert not errs, errs
        print({"parallel_streams": nthreads, "chunk_mb": chunk_mb,
               "aggregate_gb_per_s": round(size * reps * nthreads / dt / 1e9,
                                           2)})
    finally:
        m0.stop()
        m1.stop()
        driver.stop()


if __name__ == "__main__":
    for chunk in (1, 4, 16):
        run(chunk_mb=chunk)
    run(chunk_mb=4, compress=True, size_mb=128)
    run_parallel(nthreads=4)
    run_parallel(nthreads=8)
