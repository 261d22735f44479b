"""Slot-recycling generation protocol, tested deterministically on CPU
with a FAKE native layer: slots recycle with advancing generations and
resolve() re-checks the published generation per fetch, reopening (and
closing the stale mapping) when a slot was reused — the correctness
guard that lets a long-running executor's 1024-slot table never grow."""

import pytest

from sparkrdma_amd.conf import ShuffleConf
from sparkrdma_amd.map_output import make_key
from sparkrdma_amd.segments import (HostSegment, META_SEGMENT_ID,
                                    SegmentRegistry, segment_path)


class FakeHs:
    """Stand-in for the hipshuffle extension: slabs are fake pointers,
    ipc handles encode the slab id."""

    def __init__(self):
        self.next_id = 1
        self.opened = []
        self.closed = []

    def set_device(self, d):
        pass

    def slab_alloc(self, size):
        sid = self.next_id
        self.next_id += 1
        return sid

    def slab_base(self, sid):
        return 0x1000 * sid

    def slab_handle(self, sid):
        return sid.to_bytes(8, "little") + b"\0" * 56

    def slab_free(self, sid):
        pass

    def ipc_open(self, handle):
        sid = int.from_bytes(handle[:8], "little")
        self.opened.append(sid)
        return 0x9000_0000 + 0x1000 * sid

    def ipc_close(self, base):
        self.closed.append(base)


@pytest.fixture
def planes(tmp_path, monkeypatch):
    import sparkrdma_amd.ops as ops
    fake = FakeHs()
    monkeypatch.setattr(ops, "load", lambda *a, **k: fake)
    from sparkrdma_amd.gpu_plane import GpuDataPlane
    conf = ShuffleConf(shm_dir=str(tmp_path), hbm_pool_size=64 << 20,
                       hbm_slab_size=16 << 20)
    app = "testapp"
    owner_meta = HostSegment(segment_path(str(tmp_path), app, 0,
                                          META_SEGMENT_ID), 1 << 20)
    reg0 = SegmentRegistry(str(tmp_path), app)
    owner = GpuDataPlane(conf, executor_id=0, meta_segment=owner_meta,
                         registry=reg0, device=0)
    reader_meta = HostSegment(segment_path(str(tmp_path), app, 1,
                                           META_SEGMENT_ID), 1 << 20)
    reg1 = SegmentRegistry(str(tmp_path), app)
    reader = GpuDataPlane(conf, executor_id=1, meta_segment=reader_meta,
                          registry=reg1, device=0)
    yield owner, reader, fake
    reg0.close()
    reg1.close()
    owner_meta.close()
    reader_meta.close()


def test_resolve_reopens_recycled_slot(planes):
    owner, reader, fake = planes
    seg1 = owner._alloc_slab(16 << 20)        # slot 0, gen 1
    key = make_key(0, seg1)
    base1 = reader.resolve(key)
    assert fake.opened == [1]
    # cache hit while the generation is unchanged: no reopen
    assert reader.resolve(key) == base1
    assert fake.opened == [1]
    # recycle the slot: free + re-alloc gets the SAME slot, gen 2
    owner._free_slab(seg1)
    seg2 = owner._alloc_slab(16 << 20)
    assert seg2 == seg1, "slot was not recycled"
    assert owner._slab_gens[seg1 & 0x7FFF] == 2
    base2 = reader.resolve(key)
    assert base2 != base1, "stale mapping served for a recycled slot"
    assert fake.opened == [1, 2]
    assert fake.closed == [base1], "stale mapping was not closed"


def test_resolve_unpublished_slot_raises(planes):
    owner, reader, fake = planes
    seg = owner._alloc_slab(16 << 20)
    owner._free_slab(seg)                     # entry zeroed
    with pytest.raises(RuntimeError):
        reader.resolve(make_key(0, seg))


def test_slot_count_bounded_over_many_cycles(planes):
    owner, _reader, _fake = planes
    for _ in range(50):
        seg = owner._alloc_slab(16 << 20)
        owner._free_slab(seg)
    assert owner._next_slot == 1, "slots leaked instead of recycling"
    assert owner._slab_gens[0] == 50 % 0x10000


def test_importer_tick_updates_recycled_mapping(planes, tmp_path,
                                                monkeypatch):
    owner, reader, fake = planes
    # _import_published_slabs is gated on CUDA presence in production;
    # the fake hs stands in for it here
    import torch
    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)

    seg = owner._alloc_slab(16 << 20)
    key = make_key(0, seg)
    base1 = reader.resolve(key)
    owner._free_slab(seg)
    owner._alloc_slab(16 << 20)               # same slot, new gen
    reader._import_published_slabs(0)
    assert reader._peer_bases[key][0] != base1
    assert fake.closed == [base1]
