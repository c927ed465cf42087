"""Client connection to a compute node.

Capability-parity with the reference's `Connection`
(/root/reference/distllm/control_center.py:88-249): chunked uploads with
checksums and retry, slice listing/loading, status, forward propagation.
Unlike the reference (a fresh TCP connection per request) one
persistent socket is kept per node with transparent reconnect.
"""
from __future__ import annotations

import hashlib
import json
import os
import socket
from typing import List, Optional, Tuple

import numpy as np

from . import protocol as P

CHUNK = 1 << 20  # 1 MiB upload chunks
RETRIES = 3


class OperationFailedError(Exception):
    def __init__(self, resp: P.ResponseError):
        super().__init__(f"{resp.operation}: {resp.error} — {resp.description}")
        self.resp = resp


class Connection:
    def __init__(self, host: str, port: int, timeout: float = 600.0):
        self.addr = (host, port)
        self.timeout = timeout
        self._sock: Optional[socket.socket] = None

    # ------------------------------------------------------------- plumbing

    def _connect(self) -> socket.socket:
        if self._sock is None:
            s = socket.create_connection(self.addr, timeout=self.timeout)
            s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            self._sock = s
        return self._sock

    def close(self) -> None:
        if self._sock is not None:
            try:
                self._sock.close()
            finally:
                self._sock = None

    def _rpc(self, msg: P.Message, retries: int = 1) -> P.Message:
        last: Exception | None = None
        for _ in range(max(retries, 1)):
            try:
                s = self._connect()
                P.send_message(s, msg)
                resp = P.receive_message(s)
            except (ConnectionError, OSError, P.IntegrityError) as e:
                last = e
                self.close()
                continue
            if isinstance(resp, P.ResponseError):
                raise OperationFailedError(resp)
            return resp
        raise ConnectionError(f"rpc to {self.addr} failed: {last}")

    # ------------------------------------------------------------------ api

    def get_status(self) -> P.ResponseStatus:
        return self._rpc(P.RequestStatus(), retries=RETRIES)

    def list_slices(self) -> List[dict]:
        resp = self._rpc(P.RequestListSlices(), retries=RETRIES)
        return json.loads(resp.slices)

    def load_slice(self, name: str) -> P.ResponseLoadSlice:
        return self._rpc(P.RequestLoadSlice(name=name))

    def clear_context(self) -> None:
        self._rpc(P.RequestClearContext())

    def push_file(self, path: str, kind: str, metadata: dict,
                  progress=None) -> P.ResponseUploadEnd:
        size = os.path.getsize(path)
        meta = dict(metadata)
        meta.setdefault("name", os.path.basename(path))
        begin = self._rpc(P.RequestUploadBegin(kind=kind,
                                               metadata=json.dumps(meta)))
        uid = begin.upload_id
        hasher = hashlib.sha256()
        sent = 0
        with open(path, "rb") as f:
            while True:
                chunk = f.read(CHUNK)
                if not chunk:
                    break
                hasher.update(chunk)
                for attempt in range(RETRIES):
                    try:
                        self._rpc(P.RequestUploadPart(upload_id=uid,
                                                      data=chunk))
                        break
                    except (ConnectionError, P.IntegrityError):
                        if attempt == RETRIES - 1:
                            raise
                sent += len(chunk)
                if progress:
                    progress(sent, size)
        return self._rpc(P.RequestUploadEnd(upload_id=uid, total_size=size,
                                            checksum=hasher.hexdigest()))

    def push_slice(self, path: str, metadata: dict,
                   progress=None) -> P.ResponseUploadEnd:
        return self.push_file(path, "slice", metadata, progress)

    def propagate_forward(self, x: np.ndarray,
                          start_pos: int) -> np.ndarray:
        x = np.ascontiguousarray(x, dtype=np.float32)
        if x.ndim != 2:
            raise ValueError("activations must be [T, E]")
        resp = self._rpc(P.RequestPropagateForward(
            values=x.reshape(-1), axis0=x.shape[0], axis1=x.shape[1],
            start_pos=start_pos))
        if (resp.axis0, resp.axis1) != x.shape:
            raise OperationFailedError(P.ResponseError(
                operation="propagate_forward", error="shape_mismatch",
                description=f"got {(resp.axis0, resp.axis1)}, "
                            f"sent {x.shape}"))
        return resp.values.reshape(resp.axis0, resp.axis1)


def parse_address(addr: str) -> Tuple[str, int]:
    host, port = addr.rsplit(":", 1)
    return host, int(port)
