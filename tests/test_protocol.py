"""Wire protocol: byte-exact framing, codec round-trips, integrity."""
import hashlib
import struct

import numpy as np
import pytest

from distributedllm_amd.cluster import protocol as P


class FakeSocket:
    """In-memory socket with configurable read chunk sizes."""

    def __init__(self, chunk=7):
        self.buf = bytearray()
        self.chunk = chunk
        self.pos = 0

    def sendall(self, data):
        self.buf += data

    def recv(self, n):
        take = min(n, self.chunk, len(self.buf) - self.pos)
        out = bytes(self.buf[self.pos:self.pos + take])
        self.pos += take
        return out


MESSAGES = [
    P.RequestStatus(),
    P.RequestLoadSlice(name="slice_0_12.bin"),
    P.RequestUploadBegin(kind="slice", metadata='{"name": "x"}'),
    P.RequestUploadPart(upload_id=3, data=b"\x00\x01\xffbytes"),
    P.RequestUploadEnd(upload_id=3, total_size=7, checksum="ab" * 32),
    P.RequestPropagateForward(values=np.arange(6, dtype=np.float32),
                              axis0=2, axis1=3, start_pos=5),
    P.ResponseStatus(status="up", model="m", first_layer=2, n_layers=5,
                     device="cuda"),
    P.ResponsePropagateForward(values=np.zeros(4, dtype=np.float32),
                               axis0=1, axis1=4),
    P.ResponseError(operation="load_slice", error="slice_not_found",
                    description="nope"),
]


@pytest.mark.parametrize("msg", MESSAGES, ids=lambda m: m.msg_name())
def test_roundtrip(msg):
    sock = FakeSocket(chunk=5)
    P.send_message(sock, msg)
    got = P.receive_message(sock)
    assert type(got) is type(msg)
    for f in msg.__dataclass_fields__:
        a, b = getattr(msg, f), getattr(got, f)
        if isinstance(a, np.ndarray):
            assert np.array_equal(a, b)
        else:
            assert a == b


def test_frame_layout():
    msg = P.RequestStatus()
    sock = FakeSocket()
    P.send_message(sock, msg)
    raw = bytes(sock.buf)
    (length,) = struct.unpack_from("<I", raw, 0)
    digest = raw[4:36]
    payload = raw[36:]
    assert len(payload) == length
    assert hashlib.sha256(payload).digest() == digest
    # payload: name_len, name, n_fields
    assert payload[0] == len(b"request_status")
    assert payload[1:15] == b"request_status"
    assert struct.unpack_from("<H", payload, 15)[0] == 0


def test_integrity_error():
    msg = P.RequestLoadSlice(name="x")
    sock = FakeSocket()
    P.send_message(sock, msg)
    sock.buf[-1] ^= 0xFF  # corrupt payload
    with pytest.raises(P.IntegrityError):
        P.receive_message(sock)


def test_truncation_raises():
    msg = P.RequestLoadSlice(name="x")
    sock = FakeSocket()
    P.send_message(sock, msg)
    del sock.buf[-3:]
    with pytest.raises(ConnectionError):
        P.receive_message(sock)


def test_float_array_is_raw_le():
    v = np.array([1.0, -2.5], dtype=np.float32)
    enc = P.RequestPropagateForward(values=v, axis0=1, axis1=2,
                                    start_pos=0).encode()
    assert v.astype("<f4").tobytes() in enc


def test_unknown_message_rejected():
    with pytest.raises(P.ProtocolError):
        P.Message.decode(b"\x03xyz\x00\x00")


def test_garbage_bytes_rejected():
    """Corrupt/hostile frames raise typed protocol errors, never crash:
    bad magic lengths, truncation, checksum mismatch, unknown names."""
    import random
    rng = random.Random(0)
    # oversized length field
    sock = FakeSocket(chunk=64)
    sock.buf += struct.pack("<I", P.MAX_PAYLOAD + 1) + b"\x00" * 32
    with pytest.raises(P.ProtocolError):
        P.receive_message(sock)
    # random garbage payloads with VALID framing -> decode errors, typed
    for _ in range(50):
        payload = bytes(rng.randrange(256) for _ in range(rng.randrange(1, 60)))
        sock = FakeSocket(chunk=7)
        digest = hashlib.sha256(payload).digest()
        sock.buf += struct.pack("<I", len(payload)) + digest + payload
        with pytest.raises(P.ProtocolError):
            P.receive_message(sock)
    # checksum mismatch
    msg = P.RequestStatus().encode()
    sock = FakeSocket(chunk=16)
    sock.buf += struct.pack("<I", len(msg)) + b"\x00" * 32 + msg
    with pytest.raises(P.IntegrityError):
        P.receive_message(sock)
