"""NAT-traversal proxy relay + node reverse-connect.

Capability-parity with the reference's proxy
(/root/reference/distllm/proxy_node.py:12-81 and the reverse-connect
handshake at compute_node/serve.py:26-56): a compute node behind NAT dials
OUT to the proxy's node port and announces itself; clients connect to the
proxy's client port and speak the normal node protocol; the proxy bridges
the two. Re-designed rather than translated:

* frames are relayed verbatim at the framing layer (length ‖ sha256 ‖
  payload) — the proxy never decodes message bodies, so tensor hops are
  not re-parsed in the middle (the reference decodes and re-encodes every
  message through its Queue(1) pairs),
* one lock serializes client round-trips over the single node socket
  (the reference serializes via the two size-1 queues),
* a dropped node is detected on the next round-trip and reported to the
  client as a ``ResponseError`` instead of a hang; the node auto-redials.
"""
from __future__ import annotations

import socket
import socketserver
import struct
import threading
import time
from typing import Optional

from . import protocol as P


def read_frame(sock: socket.socket) -> bytes:
    """One raw protocol frame (header + payload), unverified."""
    header = P._recv_exact(sock, 4 + 32)
    (length,) = struct.unpack_from("<I", header, 0)
    if length > P.MAX_PAYLOAD:
        raise P.ProtocolError(f"payload too large: {length}")
    return header + P._recv_exact(sock, length)


class ProxyServer:
    """Bridges one reverse-connected compute node to many clients."""

    def __init__(self, host: str, client_port: int, node_port: int):
        self.host = host
        self._node_sock: Optional[socket.socket] = None
        self._node_name = ""
        self._node_lock = threading.Lock()
        self._stop = threading.Event()

        proxy = self

        class _NodeHandler(socketserver.BaseRequestHandler):
            def handle(self):
                proxy._serve_node(self.request)

        class _ClientHandler(socketserver.BaseRequestHandler):
            def handle(self):
                proxy._serve_client(self.request)

        class _Srv(socketserver.ThreadingTCPServer):
            allow_reuse_address = True
            daemon_threads = True

        self._node_srv = _Srv((host, node_port), _NodeHandler)
        self._client_srv = _Srv((host, client_port), _ClientHandler)

    @property
    def node_port(self) -> int:
        return self._node_srv.server_address[1]

    @property
    def client_port(self) -> int:
        return self._client_srv.server_address[1]

    # ------------------------------------------------------------- node side

    def _serve_node(self, sock: socket.socket) -> None:
        try:
            greeting = P.receive_message(sock)
        except (P.ProtocolError, ConnectionError, OSError):
            sock.close()
            return
        if not isinstance(greeting, P.RequestGreeting):
            P.send_message(sock, P.ResponseError(
                operation="greeting", error="bad_handshake",
                description="expected request_greeting"))
            sock.close()
            return
        # a stalled-but-connected node must not hold the roundtrip lock
        # forever: bound every node read/write so _roundtrip converts a
        # stall into node_lost (socket.timeout is an OSError subclass,
        # so the except path below already covers it)
        sock.settimeout(120.0)
        with self._node_lock:
            old = self._node_sock
            self._node_sock = sock
            self._node_name = greeting.name
            if old is not None:
                try:
                    old.close()
                except OSError:
                    pass
        P.send_message(sock, P.ResponseGreeting(status="ok"))
        # The node socket is driven by client threads; this handler thread
        # just parks until the server shuts down or the socket is replaced.
        while not self._stop.is_set():
            with self._node_lock:
                if self._node_sock is not sock:
                    return
            time.sleep(0.05)

    # ----------------------------------------------------------- client side

    def _serve_client(self, sock: socket.socket) -> None:
        sock.settimeout(600.0)
        try:
            while not self._stop.is_set():
                try:
                    frame = read_frame(sock)
                except (ConnectionError, socket.timeout, P.ProtocolError):
                    return
                reply = self._roundtrip(frame)
                sock.sendall(reply)
        finally:
            try:
                sock.close()
            except OSError:
                pass

    def _roundtrip(self, frame: bytes) -> bytes:
        with self._node_lock:
            node = self._node_sock
            if node is None:
                return _error_frame("proxy", "no_node",
                                    "no compute node connected to proxy")
            try:
                node.sendall(frame)
                return read_frame(node)
            except (ConnectionError, OSError, P.ProtocolError) as e:
                try:
                    node.close()
                except OSError:
                    pass
                self._node_sock = None
                return _error_frame("proxy", "node_lost",
                                    f"{type(e).__name__}: {e}")

    # ------------------------------------------------------------- lifecycle

    def start(self) -> None:
        for srv in (self._node_srv, self._client_srv):
            t = threading.Thread(target=srv.serve_forever, daemon=True)
            t.start()

    def shutdown(self) -> None:
        self._stop.set()
        self._node_srv.shutdown()
        self._client_srv.shutdown()
        self._node_srv.server_close()
        self._client_srv.server_close()
        with self._node_lock:
            if self._node_sock is not None:
                try:
                    self._node_sock.close()
                except OSError:
                    pass
                self._node_sock = None


def _error_frame(operation: str, error: str, description: str) -> bytes:
    msg = P.ResponseError(operation=operation, error=error,
                          description=description)
    payload = msg.encode()
    import hashlib
    return (struct.pack("<I", len(payload))
            + hashlib.sha256(payload).digest() + payload)


def run_proxy(host: str, client_port: int, node_port: int) -> ProxyServer:
    srv = ProxyServer(host, client_port, node_port)
    srv.start()
    print(f"[proxy] clients on {host}:{srv.client_port}, "
          f"nodes on {host}:{srv.node_port}")
    return srv


# --------------------------------------------------------- node reverse mode

def connect_then_serve(proxy_host: str, proxy_port: int, state,
                       name: str = "node", stop: Optional[threading.Event] = None,
                       redial_delay: float = 1.0) -> None:
    """Dial out to the proxy, announce, then serve requests over the one
    outbound socket; redial on disconnect (reference serve.py:35-46)."""
    stop = stop or threading.Event()
    while not stop.is_set():
        try:
            sock = socket.create_connection((proxy_host, proxy_port),
                                            timeout=600.0)
        except OSError:
            if stop.wait(redial_delay):
                return
            continue
        try:
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            P.send_message(sock, P.RequestGreeting(name=name))
            ack = P.receive_message(sock)
            if not isinstance(ack, P.ResponseGreeting):
                raise P.ProtocolError("proxy refused handshake")
            while not stop.is_set():
                msg = P.receive_message(sock)
                resp = state.handle(msg)
                P.send_message(sock, resp)
        except (ConnectionError, OSError, P.ProtocolError):
            pass
        finally:
            try:
                sock.close()
            except OSError:
                pass
        if stop.wait(redial_delay):
            return
