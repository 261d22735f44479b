"""SPMD execution engine: one executor process per GPU.

The reference has no runtime of its own — Spark is the runtime above the
plugin. The rebuild needs a minimal one so the workloads
(TeraSort, PageRank, SQL join) can run standalone: N identical processes
(torchrun / multiprocessing), each owning one GPU; process 0 additionally
hosts the Driver registry thread. Stage boundaries are driver barriers.

Environment contract (torchrun-compatible): RANK, WORLD_SIZE, MASTER_ADDR,
MASTER_PORT select identity; a fixed offset of MASTER_PORT is used for the
shuffle driver's own listener so both can coexist.
"""

from __future__ import annotations

import logging
import os
from typing import Optional

from .conf import ShuffleConf
from .driver import Driver
from .manager import ShuffleManager

log = logging.getLogger(__name__)

DRIVER_PORT_OFFSET = 71  # driver listens on MASTER_PORT + this


class Engine:
    """Per-process handle: driver (rank 0 only) + executor-side manager."""

    def __init__(self, conf: Optional[ShuffleConf] = None,
                 rank: Optional[int] = None, world_size: Optional[int] = None,
                 driver_port: Optional[int] = None):
        self.conf = conf or ShuffleConf()
        self.rank = int(os.environ.get("RANK", 0)) if rank is None else rank
        self.world_size = (int(os.environ.get("WORLD_SIZE", 1))
                           if world_size is None else world_size)
        if driver_port is None:
            base = int(os.environ.get("MASTER_PORT", 29500))
            driver_port = base + DRIVER_PORT_OFFSET
        host = os.environ.get("MASTER_ADDR", self.conf.driver_host)
        # container hostnames may not resolve; keep loopback default
        self.conf.driver_host = host if host not in ("localhost",) else "127.0.0.1"
        self.driver: Optional[Driver] = None
        if self.rank == 0:
            dconf = self.conf
            dconf.driver_port = driver_port
            self.driver = Driver(dconf)
            driver_port = self.driver.port  # resolves port-0 binds
        elif driver_port == 0:
            raise ValueError("non-zero driver_port required on ranks > 0")
        self.manager = ShuffleManager(self.conf, executor_id=self.rank,
                                      driver_port=driver_port)
        # wait until every executor is announced before any stage runs
        import time
        deadline = time.monotonic() + 60
        while len(self.manager._members) < self.world_size:
            if time.monotonic() > deadline:
                raise TimeoutError(
                    f"only {len(self.manager._members)}/{self.world_size} "
                    "executors announced")
            time.sleep(0.005)

    # convenience passthroughs -----------------------------------------

    def register_shuffle(self, num_maps: int, num_partitions: int):
        """SPMD-collective: rank 0 registers with the driver; every rank
        derives the same handle (ids are sequential, the table path is
        deterministic in (app_id, shuffle_id)) — the analog of Spark
        broadcasting the serialized handle to executors."""
        from .manager import ShuffleHandle
        from .segments import driver_table_path
        sid = getattr(self, "_shuffle_seq", 0)
        self._shuffle_seq = sid + 1
        if self.rank == 0:
            h = self.manager.register_shuffle(num_maps, num_partitions)
            if h.shuffle_id != sid:
                raise RuntimeError(
                    f"shuffle id drift: driver={h.shuffle_id} local={sid} "
                    "(register_shuffle must go through Engine)")
        self.barrier()  # driver table exists before anyone writes it
        if self.rank != 0:
            # verify the derived id against the authoritative registry so
            # out-of-band registrations desynchronize LOUDLY on every rank
            # (only rank 0 asserted)
            nm, np_, _uri = self.manager.lookup_shuffle(sid)
            if (nm, np_) != (num_maps, num_partitions):
                raise RuntimeError(
                    f"shuffle id drift on rank {self.rank}: id {sid} is "
                    f"({nm} maps, {np_} parts) at the driver, expected "
                    f"({num_maps}, {num_partitions})")
        return ShuffleHandle(sid, num_maps, num_partitions,
                             driver_table_path(self.conf.shm_dir,
                                               self.manager.app_id, sid))

    def unregister_shuffle(self, handle) -> None:
        self.barrier()  # every reader done with served blocks
        self.manager.unregister_shuffle(handle.shuffle_id,
                                        notify_driver=self.rank == 0)

    def barrier(self) -> None:
        self.manager.barrier()

    def shutdown(self) -> None:
        self.manager.stop()
        if self.driver is not None:
            self.driver.stop()

    # context-manager form: `with Engine(conf) as eng: ...` tears the
    # control plane down on any exit path
    def __enter__(self) -> "Engine":
        return self

    def __exit__(self, exc_type, exc, tb) -> None:
        self.shutdown()
