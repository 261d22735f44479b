"""RPC framing + segmentation codec contract (reference RdmaRpcMsg.scala:
40-78: 4B len + 4B type header, messages segmented to recvWrSize)."""

import pytest

from sparkrdma_amd import rpc


def test_hello_roundtrip():
    info = rpc.ExecutorInfo(3, "10.0.0.7", 1234, 5, "/dev/shm/x_e3_s1")
    body = rpc.pack_hello(info)
    assert rpc.unpack_hello(body) == info


def test_announce_roundtrip():
    members = [rpc.ExecutorInfo(i, f"host{i}", 100 + i, i % 8, f"seg{i}")
               for i in range(8)]
    body = rpc.pack_announce("app42", members)
    app_id, out = rpc.unpack_announce(body)
    assert app_id == "app42"
    assert out == members


def test_handle_roundtrip():
    body = rpc.pack_handle(7, 64, 512, "/dev/shm/drv_sh7", 0)
    assert rpc.unpack_handle(body) == (7, 64, 512, "/dev/shm/drv_sh7", 0)


def test_single_frame():
    frames = rpc.encode_frames(rpc.MSG_HELLO, b"abc", max_frame=4096)
    assert len(frames) == 1
    dec = rpc.FrameDecoder()
    msgs = dec.feed(frames[0])
    assert msgs == [(rpc.MSG_HELLO, b"abc")]


def test_segmentation_and_reassembly():
    body = bytes(range(256)) * 40  # 10240 bytes
    frames = rpc.encode_frames(rpc.MSG_ANNOUNCE, body, max_frame=4096)
    assert len(frames) == 3  # ceil(10240 / (4096-12))
    assert all(len(f) <= 4096 for f in frames)
    dec = rpc.FrameDecoder()
    # drip-feed byte by byte across frame boundaries
    msgs = []
    stream = b"".join(frames)
    for i in range(0, len(stream), 1000):
        msgs.extend(dec.feed(stream[i:i + 1000]))
    assert msgs == [(rpc.MSG_ANNOUNCE, body)]


def test_back_to_back_messages():
    f1 = rpc.encode_frames(rpc.MSG_HELLO, b"one", 4096)
    f2 = rpc.encode_frames(rpc.MSG_BYE, b"two", 4096)
    dec = rpc.FrameDecoder()
    msgs = dec.feed(b"".join(f1 + f2))
    assert msgs == [(rpc.MSG_HELLO, b"one"), (rpc.MSG_BYE, b"two")]


def test_empty_body():
    frames = rpc.encode_frames(rpc.MSG_BARRIER, b"", 4096)
    dec = rpc.FrameDecoder()
    assert dec.feed(frames[0]) == [(rpc.MSG_BARRIER, b"")]


def test_max_frame_too_small():
    with pytest.raises(ValueError):
        rpc.encode_frames(0, b"x", max_frame=8)


def test_decoder_fuzz_random_splits():
    """A stream of many segmented messages fed to the decoder in random-
    sized slices (1..7000 B) must reassemble every message, in order,
    regardless of where TCP read boundaries fall."""
    import random
    rng = random.Random(1234)
    msgs = []
    stream = bytearray()
    for i in range(60):
        body = bytes(rng.getrandbits(8) for _ in range(rng.randrange(0, 5000)))
        mtype = rng.randrange(0, 13)
        msgs.append((mtype, body))
        for f in rpc.encode_frames(mtype, body, rng.choice((64, 256, 4096))):
            stream.extend(f)
    dec = rpc.FrameDecoder()
    got = []
    off = 0
    while off < len(stream):
        n = rng.randrange(1, 7000)
        got.extend(dec.feed(bytes(stream[off:off + n])))
        off += n
    assert got == msgs
