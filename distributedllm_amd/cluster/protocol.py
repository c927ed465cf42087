"""Control-plane wire protocol.

Same capability set as the reference's TCP protocol
(/root/reference/distllm/protocol.py: 17 message types framed as
``len ‖ sha256 ‖ name ‖ typed body``) but a clean re-design:

* frame: ``u32 payload_len ‖ 32-byte raw sha256(payload) ‖ payload``
  (binary digest, not the reference's 64-char hexdigest).
* payload: ``u8 name_len ‖ name ‖ u16 n_fields ‖ fields``;
  field = ``u8 key_len ‖ key ‖ u8 tag ‖ value``.
* value tags: ``i`` s64, ``f`` f64, ``s`` utf-8 (u32 len), ``b`` blob
  (u32 len), ``a`` f32 array (u32 count ‖ raw LE floats).

Tensors travel as raw f32 arrays — never per-element boxed values (the
reference's list-of-floats encoding, utils.py:72-94, is the known perf
mistake). In the MI355X deployment this TCP plane carries only
control + provisioning; activations move over RCCL (parallel/pipeline.py).
On CPU-only clusters it is also a functional data plane.
"""
from __future__ import annotations

import hashlib
import socket
import struct
from dataclasses import dataclass, fields as dc_fields
from typing import Dict, Type

import numpy as np

MAX_PAYLOAD = 1 << 30


class ProtocolError(Exception):
    pass


class IntegrityError(ProtocolError):
    pass


# ----------------------------------------------------------------- codec

def _enc_value(v) -> bytes:
    if isinstance(v, bool):
        return b"i" + struct.pack("<q", int(v))
    if isinstance(v, int):
        return b"i" + struct.pack("<q", v)
    if isinstance(v, float):
        return b"f" + struct.pack("<d", v)
    if isinstance(v, str):
        raw = v.encode("utf-8")
        return b"s" + struct.pack("<I", len(raw)) + raw
    if isinstance(v, (bytes, bytearray)):
        return b"b" + struct.pack("<I", len(v)) + bytes(v)
    if isinstance(v, np.ndarray):
        a = np.ascontiguousarray(v, dtype="<f4")
        return b"a" + struct.pack("<I", a.size) + a.tobytes()
    raise ProtocolError(f"cannot encode value of type {type(v)}")


def _dec_value(data: bytes, off: int):
    tag = data[off:off + 1]
    off += 1
    if tag == b"i":
        (v,) = struct.unpack_from("<q", data, off)
        return v, off + 8
    if tag == b"f":
        (v,) = struct.unpack_from("<d", data, off)
        return v, off + 8
    if tag == b"s":
        (n,) = struct.unpack_from("<I", data, off)
        off += 4
        return data[off:off + n].decode("utf-8"), off + n
    if tag == b"b":
        (n,) = struct.unpack_from("<I", data, off)
        off += 4
        return bytes(data[off:off + n]), off + n
    if tag == b"a":
        (n,) = struct.unpack_from("<I", data, off)
        off += 4
        v = np.frombuffer(data, dtype="<f4", count=n, offset=off).copy()
        return v, off + 4 * n
    raise ProtocolError(f"unknown value tag {tag!r}")


# --------------------------------------------------------------- messages

_REGISTRY: Dict[str, Type["Message"]] = {}


@dataclass
class Message:
    """Base: subclasses are dataclasses auto-registered by ``msg`` name."""

    def __init_subclass__(cls, **kw):
        super().__init_subclass__(**kw)
        _REGISTRY[cls.msg_name()] = cls

    @classmethod
    def msg_name(cls) -> str:
        # CamelCase -> snake_case
        out = []
        for i, ch in enumerate(cls.__name__):
            if ch.isupper() and i:
                out.append("_")
            out.append(ch.lower())
        return "".join(out)

    def encode(self) -> bytes:
        name = self.msg_name().encode("ascii")
        flds = dc_fields(self)
        out = [struct.pack("<B", len(name)), name,
               struct.pack("<H", len(flds))]
        for f in flds:
            key = f.name.encode("ascii")
            out.append(struct.pack("<B", len(key)))
            out.append(key)
            out.append(_enc_value(getattr(self, f.name)))
        return b"".join(out)

    @staticmethod
    def decode(payload: bytes) -> "Message":
        off = 0
        nlen = payload[0]
        off = 1
        name = payload[off:off + nlen].decode("ascii")
        off += nlen
        (nf,) = struct.unpack_from("<H", payload, off)
        off += 2
        kv = {}
        for _ in range(nf):
            klen = payload[off]
            off += 1
            key = payload[off:off + klen].decode("ascii")
            off += klen
            val, off = _dec_value(payload, off)
            kv[key] = val
        cls = _REGISTRY.get(name)
        if cls is None:
            raise ProtocolError(f"unknown message {name!r}")
        return cls(**kv)


# requests
@dataclass
class RequestStatus(Message):
    pass


@dataclass
class RequestListSlices(Message):
    pass


@dataclass
class RequestLoadSlice(Message):
    name: str = ""


@dataclass
class RequestUploadBegin(Message):
    kind: str = "slice"     # "slice" | "file"
    metadata: str = "{}"    # JSON


@dataclass
class RequestUploadPart(Message):
    upload_id: int = 0
    data: bytes = b""


@dataclass
class RequestUploadEnd(Message):
    upload_id: int = 0
    total_size: int = 0
    checksum: str = ""      # sha256 hexdigest of the whole file


@dataclass
class RequestPropagateForward(Message):
    values: np.ndarray = None
    axis0: int = 0
    axis1: int = 0
    start_pos: int = 0      # n_past of the first token


@dataclass
class RequestClearContext(Message):
    pass


@dataclass
class RequestGreeting(Message):
    """Reverse-connect handshake: a node announcing itself to a proxy
    (reference serve.py:49-56)."""
    name: str = "node"


# responses
@dataclass
class ResponseGreeting(Message):
    status: str = "ok"

@dataclass
class ResponseStatus(Message):
    status: str = "up"
    model: str = ""
    first_layer: int = -1
    n_layers: int = 0
    device: str = "cpu"


@dataclass
class ResponseListSlices(Message):
    slices: str = "[]"      # JSON list of {name, metadata}


@dataclass
class ResponseLoadSlice(Message):
    name: str = ""
    first_layer: int = -1
    n_layers: int = 0


@dataclass
class ResponseUploadBegin(Message):
    upload_id: int = 0


@dataclass
class ResponseUploadPart(Message):
    total_received: int = 0


@dataclass
class ResponseUploadEnd(Message):
    name: str = ""
    total_size: int = 0


@dataclass
class ResponsePropagateForward(Message):
    values: np.ndarray = None
    axis0: int = 0
    axis1: int = 0


@dataclass
class ResponseClearContext(Message):
    pass


@dataclass
class ResponseError(Message):
    operation: str = ""
    error: str = ""
    description: str = ""


# ---------------------------------------------------------------- framing

def send_message(sock: socket.socket, msg: Message) -> None:
    payload = msg.encode()
    digest = hashlib.sha256(payload).digest()
    sock.sendall(struct.pack("<I", len(payload)) + digest + payload)


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    chunks = []
    got = 0
    while got < n:
        chunk = sock.recv(min(n - got, 1 << 20))
        if not chunk:
            raise ConnectionError("socket closed mid-message")
        chunks.append(chunk)
        got += len(chunk)
    return b"".join(chunks)


def receive_message(sock: socket.socket) -> Message:
    header = _recv_exact(sock, 4 + 32)
    (length,) = struct.unpack_from("<I", header, 0)
    if length > MAX_PAYLOAD:
        raise ProtocolError(f"payload too large: {length}")
    digest = header[4:36]
    payload = _recv_exact(sock, length)
    if hashlib.sha256(payload).digest() != digest:
        raise IntegrityError("payload sha256 mismatch")
    try:
        return Message.decode(payload)
    except ProtocolError:
        raise
    except Exception as e:  # malformed body -> typed error, never a crash
        raise ProtocolError(f"malformed message: {type(e).__name__}: {e}")
