"""Compute-node server: TCP control plane + slice execution.

Capability-parity with the reference's compute node
(/root/reference/distllm/compute_node/: serve.py ThreadingTCPServer,
routes.py 7 handlers, slices.py SliceContainer with the DummySlice test
format — SURVEY §1 L3). Differences by design:

* all shared state sits in one `NodeState` with a lock (the reference
  shares unsynchronized module singletons across threads),
* the engine is the MI355X HIP slice engine when a GPU is present, the
  fp32 torch twin otherwise; `format: "test"` loads the 2-float DummySlice
  (k·x+b) so orchestration is testable with no model at all,
* `propagate_forward` carries an explicit `start_pos` so the node holds no
  hidden n_past (stateless engines; SURVEY §3.4's n_past lives client-side).
"""
from __future__ import annotations

import json
import os
import socket
import socketserver
import threading
from typing import Optional

import numpy as np
import torch

from ..formats import ggml
from . import protocol as P
from .uploads import UploadManager, UploadError


class SliceError(Exception):
    pass


class DummySlice:
    """y = k*x + b elementwise; 8-byte file (two f32). The reference's
    multi-node-without-a-model test trick (slices.py:64-71)."""

    def __init__(self, path: str):
        raw = np.fromfile(path, dtype="<f4", count=2)
        if raw.size != 2:
            raise SliceError("dummy slice file must hold two f32 values")
        self.k, self.b = float(raw[0]), float(raw[1])
        self.name = os.path.basename(path)
        self.first_layer = 0
        self.n_layers = 0

    def forward(self, x: np.ndarray, start_pos: int) -> np.ndarray:
        return self.k * x + self.b


class EngineSlice:
    """A loaded GGML slice running on the best local engine."""

    def __init__(self, path: str, n_ctx: int = 2048, max_batch: int = 16,
                 device: Optional[str] = None):
        from ..engine import engine_for_slice
        f = ggml.GGMLFile.load(path, extended=True)
        self.engine = engine_for_slice(f, n_ctx=n_ctx, max_batch=max_batch,
                                       device=device)
        self.name = os.path.basename(path)
        self.first_layer = f.hparams.first_layer or 0
        self.n_layers = f.hparams.n_layer
        self.device = self.engine.device

    def forward(self, x: np.ndarray, start_pos: int) -> np.ndarray:
        t = torch.from_numpy(np.ascontiguousarray(x, dtype=np.float32))
        T = t.shape[0]
        if start_pos + T > self.engine.n_ctx:
            # the reference never guards n_ctx overflow —
            # here it would silently corrupt a neighboring sequence's KV
            raise SliceError(
                f"context overflow: start_pos {start_pos} + {T} tokens "
                f"exceeds n_ctx {self.engine.n_ctx}")
        dev = self.device
        pos = torch.arange(start_pos, start_pos + T, dtype=torch.int32)
        seq = torch.zeros(T, dtype=torch.int32)
        t = t.to(dev)
        y = self.engine.forward(t, pos.to(dev), seq.to(dev))
        return y.float().cpu().numpy()


class NodeState:
    def __init__(self, uploads_dir: str, device: Optional[str] = None,
                 n_ctx: int = 2048):
        self.lock = threading.Lock()
        # serializes forward passes: concurrent propagate_forward calls
        # would interleave KV-cache writes of the shared engine (the
        # reference leaves this unsynchronized — SURVEY §5.2)
        self.fwd_lock = threading.Lock()
        self.uploads = UploadManager(uploads_dir)
        self.slice = None  # DummySlice | EngineSlice
        self.device = device
        self.n_ctx = n_ctx

    # --------------------------------------------------------- handlers

    def handle(self, msg: P.Message) -> P.Message:
        name = msg.msg_name()
        handler = getattr(self, f"_on_{name}", None)
        if handler is None:
            return P.ResponseError(operation=name, error="unknown_request",
                                   description=f"no handler for {name}")
        try:
            return handler(msg)
        except UploadError as e:
            self.uploads.abort_active()
            return P.ResponseError(operation=name, error="upload_failed",
                                   description=str(e))
        except SliceError as e:
            return P.ResponseError(operation=name, error="slice_error",
                                   description=str(e))
        except FileNotFoundError as e:
            return P.ResponseError(operation=name, error="not_found",
                                   description=str(e))
        except Exception as e:  # noqa: BLE001
            return P.ResponseError(operation=name, error="internal_error",
                                   description=f"{type(e).__name__}: {e}")

    def _on_request_status(self, msg) -> P.Message:
        with self.lock:
            s = self.slice
            dev = "cuda" if torch.cuda.is_available() else "cpu"
            return P.ResponseStatus(
                status="up",
                model=s.name if s else "",
                first_layer=s.first_layer if s else -1,
                n_layers=s.n_layers if s else 0,
                device=dev)

    def _on_request_list_slices(self, msg) -> P.Message:
        recs = self.uploads.finished("slice")
        out = [{"name": os.path.basename(r.path), "metadata": r.metadata,
                "size": r.size} for r in recs]
        return P.ResponseListSlices(slices=json.dumps(out))

    def _on_request_load_slice(self, msg: P.RequestLoadSlice) -> P.Message:
        rec = self.uploads.find_slice(msg.name)
        if rec is None:
            return P.ResponseError(operation="load_slice",
                                   error="slice_not_found",
                                   description=f"no uploaded slice {msg.name!r}")
        fmt = rec.metadata.get("format", "ggml")
        try:
            if fmt == "test":
                s = DummySlice(rec.path)
            else:
                s = EngineSlice(rec.path, n_ctx=self.n_ctx,
                                device=self.device)
        except Exception as e:  # noqa: BLE001
            return P.ResponseError(operation="load_slice",
                                   error="slice_load_error",
                                   description=f"{type(e).__name__}: {e}")
        with self.lock:
            self.slice = s
        return P.ResponseLoadSlice(name=s.name, first_layer=s.first_layer,
                                   n_layers=s.n_layers)

    def _on_request_upload_begin(self, msg: P.RequestUploadBegin) -> P.Message:
        meta = json.loads(msg.metadata)
        uid = self.uploads.begin(msg.kind, meta)
        return P.ResponseUploadBegin(upload_id=uid)

    def _on_request_upload_part(self, msg: P.RequestUploadPart) -> P.Message:
        total = self.uploads.part(msg.upload_id, msg.data)
        return P.ResponseUploadPart(total_received=total)

    def _on_request_upload_end(self, msg: P.RequestUploadEnd) -> P.Message:
        rec = self.uploads.end(msg.upload_id, msg.total_size, msg.checksum)
        return P.ResponseUploadEnd(name=os.path.basename(rec.path),
                                   total_size=rec.size)

    def _on_request_propagate_forward(
            self, msg: P.RequestPropagateForward) -> P.Message:
        with self.lock:
            s = self.slice
        if s is None:
            return P.ResponseError(operation="propagate_forward",
                                   error="slice_not_loaded",
                                   description="load a slice first")
        x = msg.values.reshape(msg.axis0, msg.axis1)
        try:
            with self.fwd_lock:
                y = s.forward(x, msg.start_pos)
        except Exception as e:  # noqa: BLE001
            return P.ResponseError(operation="propagate_forward",
                                   error="neural_computation_error",
                                   description=f"{type(e).__name__}: {e}")
        return P.ResponsePropagateForward(values=y.reshape(-1),
                                          axis0=y.shape[0], axis1=y.shape[1])

    def _on_request_clear_context(self, msg) -> P.Message:
        # engines are stateless over explicit positions; nothing to reset
        return P.ResponseClearContext()


class _Handler(socketserver.BaseRequestHandler):
    def handle(self):
        state: NodeState = self.server.node_state  # type: ignore[attr-defined]
        sock: socket.socket = self.request
        sock.settimeout(600.0)
        try:
            while True:
                try:
                    msg = P.receive_message(sock)
                except (ConnectionError, socket.timeout, P.ProtocolError):
                    return
                resp = state.handle(msg)
                P.send_message(sock, resp)
        finally:
            try:
                sock.close()
            except OSError:
                pass


class NodeServer(socketserver.ThreadingTCPServer):
    allow_reuse_address = True
    daemon_threads = True

    def __init__(self, host: str, port: int, uploads_dir: str,
                 device: Optional[str] = None, n_ctx: int = 2048):
        super().__init__((host, port), _Handler)
        self.node_state = NodeState(uploads_dir, device=device, n_ctx=n_ctx)

    @property
    def port(self) -> int:
        return self.server_address[1]


def run_server(host: str, port: int, uploads_dir: str,
               device: Optional[str] = None, n_ctx: int = 2048) -> None:
    srv = NodeServer(host, port, uploads_dir, device=device, n_ctx=n_ctx)
    print(f"[node] serving on {host}:{srv.port}, uploads in {uploads_dir}")
    srv.serve_forever()
