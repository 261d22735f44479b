"""Driver-side registry — membership + per-shuffle map-output tables.

MI355X re-design of the reference driver role (RdmaShuffleManager.scala:
73-134, 168-183): the driver is a pure metadata registry and connection
introducer. It

* accepts ``Hello`` from executors and fans out ``Announce`` with the full
  membership to everyone (reference receiveListener :74-115);
* on ``registerShuffle`` allocates the per-shuffle map-output address
  table. In the reference this is a registered RDMA buffer executors
  one-sidedly WRITE into (:168-172, :384-418); here it is a /dev/shm
  segment executors mmap and write 12-byte entries into directly — the
  same one-sided property on one node;
* serves a stage barrier (the rebuild has no Spark DAG scheduler above it,
  so stage boundaries are explicit).

The announce fan-out runs on the per-connection receive thread, mirroring
the reference's constraint that listener work stays non-blocking (it runs
on the CQ poller thread there — RdmaChannel.java:731).
"""

from __future__ import annotations

import logging
import socket
import threading
import uuid
from typing import Dict, List, Optional

from .conf import ShuffleConf
from .map_output import MAP_ENTRY_SIZE
from . import rpc
from .segments import HostSegment, driver_table_path

log = logging.getLogger(__name__)


class _ShuffleMeta:
    def __init__(self, shuffle_id: int, num_maps: int, num_partitions: int,
                 table: HostSegment):
        self.shuffle_id = shuffle_id
        self.num_maps = num_maps
        self.num_partitions = num_partitions
        self.table = table


class Driver:
    """TCP registry server. Runs as a thread inside rank 0 (torchrun mode)
    or standalone."""

    def __init__(self, conf: ShuffleConf, app_id: Optional[str] = None):
        self.conf = conf
        self.app_id = app_id or uuid.uuid4().hex[:12]
        self._server = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._server.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._server.bind((conf.driver_host, conf.driver_port))
        self._server.listen(128)
        self.port = self._server.getsockname()[1]
        self._members: Dict[int, rpc.ExecutorInfo] = {}
        self._conn_exec: Dict[rpc.MsgConnection, int] = {}
        self._conns: List[rpc.MsgConnection] = []
        self._shuffles: Dict[int, _ShuffleMeta] = {}
        self._next_shuffle_id = 0
        self._lock = threading.Lock()
        self._stopped = threading.Event()
        self._barrier_waiters: List[rpc.MsgConnection] = []
        self._barrier_gen = 0
        self._accept_thread = threading.Thread(
            target=self._accept_loop, name="sparkrdma-driver-accept", daemon=True)
        self._accept_thread.start()
        log.info("driver up app_id=%s port=%d", self.app_id, self.port)

    # ------------------------------------------------------------------

    def _accept_loop(self) -> None:
        while not self._stopped.is_set():
            try:
                sock, _ = self._server.accept()
            except OSError:
                return
            conn = rpc.MsgConnection(sock, self.conf.recv_wr_size)
            with self._lock:
                self._conns.append(conn)
            threading.Thread(target=self._serve_conn, args=(conn,),
                             name="sparkrdma-driver-conn", daemon=True).start()

    def _serve_conn(self, conn: rpc.MsgConnection) -> None:
        try:
            while not self._stopped.is_set():
                msg = conn.recv_any(timeout=None)
                if msg is None:
                    break
                mtype, body = msg
                self._dispatch(conn, mtype, body)
        except (OSError, ValueError):
            pass
        finally:
            # executor loss: prune membership and re-announce, the analog
            # of SparkListenerBlockManagerRemoved pruning (reference
            # RdmaShuffleManager.scala:155-165). A clean MSG_BYE departure
            # already pruned; this then finds nothing to do.
            self._prune_executor(conn, clean=False)

    def _prune_executor(self, conn: rpc.MsgConnection, clean: bool) -> None:
        """Drop an executor (clean BYE or lost connection): prune
        membership, re-announce, and release any barrier the departure
        now satisfies (everyone still alive is already waiting)."""
        reannounce = None
        release = []
        with self._lock:
            if not clean and conn in self._conns:
                self._conns.remove(conn)
            if conn in self._barrier_waiters:
                self._barrier_waiters.remove(conn)
            exec_id = self._conn_exec.pop(conn, None)
            if exec_id is not None and not self._stopped.is_set():
                self._members.pop(exec_id, None)
                if clean:
                    log.info("executor %d detached; %d members remain",
                             exec_id, len(self._members))
                else:
                    log.warning("executor %d lost; %d members remain",
                                exec_id, len(self._members))
                reannounce = (rpc.pack_announce(
                    self.app_id, list(self._members.values())),
                    [c for c in self._conns if c is not conn])
            if len(self._barrier_waiters) >= len(self._members) > 0:
                gen = self._barrier_gen
                self._barrier_gen += 1
                release = [(c, gen) for c in self._barrier_waiters]
                self._barrier_waiters = []
        if reannounce is not None:
            payload, conns = reannounce
            for c in conns:
                try:
                    c.send(rpc.MSG_ANNOUNCE, payload)
                except OSError:
                    pass
        for c, gen in release:
            try:
                c.send(rpc.MSG_BARRIER_OK, rpc.pack_unregister(gen))
            except OSError:
                pass

    def _dispatch(self, conn: rpc.MsgConnection, mtype: int, body: bytes) -> None:
        if mtype == rpc.MSG_HELLO:
            info = rpc.unpack_hello(body)
            with self._lock:
                self._members[info.executor_id] = info
                self._conn_exec[conn] = info.executor_id
                members = list(self._members.values())
                conns = list(self._conns)
            # fan out announce to every connected executor (reference :89-112)
            payload = rpc.pack_announce(self.app_id, members)
            for c in conns:
                try:
                    c.send(rpc.MSG_ANNOUNCE, payload)
                except OSError:
                    pass
        elif mtype == rpc.MSG_REGISTER:
            _, num_maps, num_partitions = rpc.unpack_register(body)
            meta = self._register_shuffle(num_maps, num_partitions)
            conn.send(rpc.MSG_HANDLE, rpc.pack_handle(
                meta.shuffle_id, num_maps, num_partitions, meta.table.path, 0))
        elif mtype == rpc.MSG_UNREGISTER:
            sid = rpc.unpack_unregister(body)
            self.unregister_shuffle(sid)
            conn.send(rpc.MSG_HANDLE, rpc.pack_handle(sid, 0, 0, "", 0))
        elif mtype == rpc.MSG_LOOKUP:
            sid = rpc.unpack_unregister(body)
            with self._lock:
                meta = self._shuffles.get(sid)
            if meta is None:
                conn.send(rpc.MSG_ERROR, rpc.pack_unregister(sid))
            else:
                conn.send(rpc.MSG_HANDLE, rpc.pack_handle(
                    sid, meta.num_maps, meta.num_partitions,
                    meta.table.path, 0))
        elif mtype == rpc.MSG_TABLE_READ:
            # cross-host hop 1: serve the driver table over RPC (executors
            # on the driver's host read it one-sidedly via mmap instead)
            sid = rpc.unpack_unregister(body)
            with self._lock:
                meta = self._shuffles.get(sid)
            if meta is None:
                conn.send(rpc.MSG_ERROR, rpc.pack_unregister(sid))
            else:
                raw = meta.table.read(0, meta.num_maps * MAP_ENTRY_SIZE)
                conn.send(rpc.MSG_TABLE_DATA, rpc.pack_table_data(sid, raw))
        elif mtype == rpc.MSG_TABLE_WRITE:
            sid, map_id, addr, key = rpc.unpack_table_write(body)
            with self._lock:
                meta = self._shuffles.get(sid)
            if meta is None or not 0 <= map_id < meta.num_maps:
                conn.send(rpc.MSG_ERROR, rpc.pack_unregister(sid))
            else:
                import struct as _struct
                meta.table.write(map_id * MAP_ENTRY_SIZE,
                                 _struct.pack("<QI", addr, key))
                conn.send(rpc.MSG_HANDLE, rpc.pack_handle(sid, 0, 0, "", 0))
        elif mtype == rpc.MSG_BYE:
            self._prune_executor(conn, clean=True)
        elif mtype == rpc.MSG_BARRIER:
            with self._lock:
                self._barrier_waiters.append(conn)
                if len(self._barrier_waiters) >= len(self._members) > 0:
                    gen = self._barrier_gen
                    self._barrier_gen += 1
                    waiters = self._barrier_waiters
                    self._barrier_waiters = []
                    for c in waiters:
                        try:
                            c.send(rpc.MSG_BARRIER_OK, rpc.pack_unregister(gen))
                        except OSError:
                            pass
        else:
            log.warning("driver: unknown message type %d", mtype)

    # ------------------------------------------------------------------

    def _register_shuffle(self, num_maps: int, num_partitions: int) -> _ShuffleMeta:
        with self._lock:
            sid = self._next_shuffle_id
            self._next_shuffle_id += 1
            path = driver_table_path(self.conf.shm_dir, self.app_id, sid)
            table = HostSegment(path, max(num_maps * MAP_ENTRY_SIZE, 4096))
            # zero-filled => key==0 everywhere == "unpublished"
            meta = _ShuffleMeta(sid, num_maps, num_partitions, table)
            self._shuffles[sid] = meta
            return meta

    def unregister_shuffle(self, shuffle_id: int) -> None:
        with self._lock:
            meta = self._shuffles.pop(shuffle_id, None)
        if meta is not None:
            meta.table.close()
            meta.table.unlink()

    def stop(self) -> None:
        self._stopped.set()
        try:
            self._server.close()
        except OSError:
            pass
        with self._lock:
            conns = list(self._conns)
            shuffles = list(self._shuffles.values())
            self._shuffles.clear()
        for c in conns:
            c.close()
        for meta in shuffles:
            meta.table.close()
            meta.table.unlink()
