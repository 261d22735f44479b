#!/usr/bin/env python3
"""Pool-pressure soak: serve >= 200 GB of HBM map outputs from ONE GPU
(VERDICT r01 item 7 done-criterion: the 288 GB pool must actually be
servable — r01's slab table capped it at 256 GiB).

Writes N shuffles of `--gb` each WITHOUT unregistering (blocks stay alive
until unregister, the reference's liveness discipline), fetches a sample
partition range from the oldest shuffle at full pressure, then
unregisters everything and asserts the pool drains to zero.
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gb", type=float, default=20.0, help="GB per shuffle")
    ap.add_argument("--target-gb", type=float, default=180.0,
                    help="stop writing when served bytes exceed this")
    ap.add_argument("--pool-gb", type=float, default=240.0)
    args = ap.parse_args()

    import torch
    from sparkrdma_amd.conf import ShuffleConf
    from sparkrdma_amd.engine import Engine
    from sparkrdma_amd.partitioner import RangePartitioner

    conf = ShuffleConf(transport="ipc",
                       hbm_pool_size=int(args.pool_gb * (1 << 30)),
                       shuffle_write_block_size=512 << 20,
                       shuffle_read_block_size=512 << 20,
                       max_bytes_in_flight=8 << 30)
    eng = Engine(conf, rank=0, world_size=1, driver_port=0)
    R = 256
    part = RangePartitioner.uniform(R)
    n = int(args.gb * (1 << 30) / 16)
    g = torch.Generator(device="cuda").manual_seed(1)
    keys = torch.randint(-2**63, 2**63 - 1, (n,), dtype=torch.int64,
                         device="cuda", generator=g)
    vals = keys.clone()
    handles = []
    served = 0
    t0 = time.perf_counter()
    try:
        while served < args.target_gb * (1 << 30):
            h = eng.register_shuffle(1, R)
            w = eng.manager.get_writer(h, 0)
            w.write_device_batch(keys, vals)
            w.stop(True, partitioner=part)
            handles.append(h)
            served += n * 16
            st = eng.manager.gpu.pool.stats
            print(f"shuffle {h.shuffle_id}: served={served/(1<<30):.0f}GiB "
                  f"pool slabs={st.slab_count} ({st.slab_bytes>>30}GiB) "
                  f"used={st.used_bytes>>30}GiB", flush=True)
        # fetch a sample range from the OLDEST shuffle at full pressure
        reader = eng.manager.get_reader(handles[0], 0, 31)
        got = sum(len(d) for _ref, d in reader)
        torch.cuda.synchronize()
        print(f"fetched {got/(1<<30):.2f} GiB from shuffle 0 "
              f"under {served/(1<<30):.0f} GiB pressure", flush=True)
        assert got > 0
        for h in handles:
            eng.unregister_shuffle(h)
        st = eng.manager.gpu.pool.stats
        print("after unregister-all:", eng.manager.gpu.pool.format_stats(),
              flush=True)
        assert st.used_bytes == 0, st.used_bytes
        print({"soak": "ok", "served_gb": round(served / (1 << 30), 1),
               "shuffles": len(handles),
               "wall_s": round(time.perf_counter() - t0, 1)})
    finally:
        eng.shutdown()


if __name__ == "__main__":
    main()
