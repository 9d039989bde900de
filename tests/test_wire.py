import socket
import threading

import numpy as np
import pytest
import torch

from tfmesos_amd import wire


def _pair():
    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)
    out = {}

    def accept():
        out["conn"], _ = srv.accept()

    t = threading.Thread(target=accept)
    t.start()
    cli = socket.create_connection(srv.getsockname())
    t.join()
    srv.close()
    return cli, out["conn"]


def test_roundtrip_basic():
    a, b = _pair()
    secret = b"k" * 32
    wire.send_msg(a, {"x": 1, "y": [1.5, "z"], "b": b"\x00\x01"}, secret)
    msg = wire.recv_msg(b, secret)
    assert msg == {"x": 1, "y": [1.5, "z"], "b": b"\x00\x01"}
    a.close(), b.close()


def test_roundtrip_ndarray_and_bf16():
    a, b = _pair()
    arr = np.arange(12, dtype=np.float32).reshape(3, 4)
    t = torch.randn(5, 7).to(torch.bfloat16)
    wire.send_msg(a, {"arr": arr, "t": t}, b"s")
    msg = wire.recv_msg(b, b"s")
    np.testing.assert_array_equal(msg["arr"], arr)
    assert torch.equal(msg["t"], t)
    a.close(), b.close()


def test_hmac_rejects_wrong_secret():
    a, b = _pair()
    wire.send_msg(a, {"x": 1}, b"right")
    with pytest.raises(wire.AuthError):
        wire.recv_msg(b, b"wrong")
    a.close(), b.close()


def test_hmac_rejects_tampered_payload():
    a, b = _pair()
    payload = wire.pack({"x": 1})
    import hashlib
    import hmac as hmac_mod
    import struct
    tag = hmac_mod.new(b"s", payload, hashlib.sha256).digest()
    evil = bytearray(payload)
    evil[-1] ^= 0xFF
    a.sendall(struct.pack(">I", len(payload) + 32) + tag + bytes(evil))
    with pytest.raises(wire.AuthError):
        wire.recv_msg(b, b"s")
    a.close(), b.close()


def test_no_pickle_in_control_frames():
    # the reference used pickle (RCE wart, tfmesos/utils.py:6-15);
    # our frames must be msgpack — a frame containing a pickled object
    # must fail to decode, not execute
    assert b"pickle" not in wire.pack({"op": "ping"})
