"""Minimal keep-alive HTTP/1.1 server for trusted localhost clients.

``http.server.BaseHTTPRequestHandler`` spends ~300 µs/request on parsing,
logging hooks, and response assembly — at a 1,000 pods/s churn the shared
fake apiserver serves ~4 requests per pod lifecycle and becomes the
harness ceiling.  This server parses exactly what our own clients
(cluster/httpconn.py, stock ``http.client``) send — request line,
Content-Length, body — and writes one pre-assembled response buffer.

Contract: ``handler(method, path, body) -> (status, body_bytes)`` for
plain responses, or ``("stream", callback)`` where ``callback(sock)``
takes over the (chunked-capable) connection — used by the apiserver's
watch endpoint.  Exceptions from the handler produce a 500 and keep the
connection alive.

This is harness/in-process infrastructure: real deployments speak to the
real kube-apiserver; tests and bench use this server so the measured
plugin/extender path is not bounded by a slow fake.
"""

from __future__ import annotations

import logging
import socket
import threading
from typing import Callable, Optional

log = logging.getLogger(__name__)

_RESP_FMT = (
    "HTTP/1.1 %d X\r\n"
    "Content-Type: application/json\r\n"
    "Content-Length: %d\r\n"
    "\r\n"
)


class FastHTTPServer:
    def __init__(
        self,
        handler: Callable,
        port: int = 0,
        host: str = "127.0.0.1",
        ssl_context=None,   # server-side TLS (kubelet-style HTTPS endpoints)
    ):
        self._ssl_context = ssl_context
        import errno
        import time

        self.handler = handler
        self._sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        # a just-stopped server's accepted sockets can hold the port for a
        # moment; a fixed-port restart (tests, daemon restart) retries
        for attempt in range(50):
            try:
                self._sock.bind((host, port))
                break
            except OSError as e:
                if e.errno != errno.EADDRINUSE or port == 0 or attempt == 49:
                    raise
                time.sleep(0.02)
        self._sock.listen(128)
        self.port = self._sock.getsockname()[1]
        self._stop = threading.Event()
        self._conn_lock = threading.Lock()
        self._conns: set = set()
        self._accept_thread: Optional[threading.Thread] = None

    # ------------------------------------------------------------------ #
    def start(self) -> "FastHTTPServer":
        self._accept_thread = threading.Thread(
            target=self._accept_loop, name="fasthttp-accept", daemon=True
        )
        self._accept_thread.start()
        return self

    def stop(self) -> None:
        self._stop.set()
        try:
            self._sock.close()
        except OSError:
            pass
        with self._conn_lock:
            for c in self._conns:
                try:
                    c.shutdown(socket.SHUT_RDWR)
                except OSError:
                    pass
                try:
                    c.close()
                except OSError:
                    pass
            self._conns.clear()

    # ------------------------------------------------------------------ #
    def _accept_loop(self) -> None:
        while not self._stop.is_set():
            try:
                conn, _addr = self._sock.accept()
            except OSError:
                return
            conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            # (TLS handshake, if configured, happens in the per-connection
            # thread — it may block)
            with self._conn_lock:
                self._conns = {c for c in self._conns if c.fileno() != -1}
                self._conns.add(conn)
            threading.Thread(
                target=self._serve_conn, args=(conn,), daemon=True
            ).start()

    def _serve_conn(self, conn: socket.socket) -> None:
        if self._ssl_context is not None:
            try:
                conn = self._ssl_context.wrap_socket(conn, server_side=True)
            except (OSError, ValueError):
                try:
                    conn.close()
                except OSError:
                    pass
                return
        f = conn.makefile("rb", buffering=65536)
        try:
            while not self._stop.is_set():
                line = f.readline(65536)
                if not line or line in (b"\r\n", b"\n"):
                    if not line:
                        return
                    continue
                try:
                    method, path, _ver = line.split(b" ", 2)
                except ValueError:
                    return  # garbage request line: drop connection
                clen = 0
                while True:
                    h = f.readline(65536)
                    if h in (b"\r\n", b"\n", b""):
                        break
                    if h[:15].lower() == b"content-length:":
                        try:
                            clen = int(h[15:].strip())
                        except ValueError:
                            clen = 0
                body = f.read(clen) if clen > 0 else b""
                try:
                    result = self.handler(method.decode(), path.decode(), body)
                except Exception as e:  # noqa: BLE001
                    log.warning("fasthttp handler error: %s", e)
                    result = (500, b'{"message":"internal error"}')
                if result[0] == "stream":
                    conn.sendall(
                        b"HTTP/1.1 200 X\r\n"
                        b"Content-Type: application/json\r\n"
                        b"Transfer-Encoding: chunked\r\n\r\n"
                    )
                    try:
                        result[1](conn)
                    finally:
                        return  # stream owns (and ends) the connection
                status, payload = result
                conn.sendall(
                    (_RESP_FMT % (status, len(payload))).encode() + payload
                )
        except (BrokenPipeError, ConnectionResetError, OSError):
            pass
        finally:
            try:
                f.close()
            except OSError:
                pass
            try:
                conn.close()
            except OSError:
                pass
