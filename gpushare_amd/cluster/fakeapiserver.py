"""Fake kube-apiserver over real HTTP.

Wraps :class:`FakeKubeClient`'s in-memory store with the REST surface that
``RestKubeClient`` and the kubelet-pods path speak, so

- unit tests exercise the *real* HTTP client code, and
- the bench harness (bench.py) can run one shared apiserver that plugin,
  scheduler-extender, and N churn-generator ranks all talk to over
  127.0.0.1 — the same process topology as a real node.

Endpoints (subset of the k8s API the framework uses):
  GET    /api/v1/nodes/{name}
  PATCH  /api/v1/nodes/{name}/status
  GET    /api/v1/pods?fieldSelector=...
  GET    /api/v1/namespaces/{ns}/pods/{name}
  PATCH  /api/v1/namespaces/{ns}/pods/{name}
  POST   /api/v1/namespaces/{ns}/pods          (bench: pod creation)
  DELETE /api/v1/namespaces/{ns}/pods/{name}   (bench: pod deletion)
  GET    /pods/                                 (kubelet read-only view)
"""

from __future__ import annotations

import json
import threading
from http.server import BaseHTTPRequestHandler
from typing import Optional
from urllib.parse import parse_qs, urlparse

from .httpconn import TrackedThreadingHTTPServer
from .kubeclient import FakeKubeClient, KubeError


class _Handler(BaseHTTPRequestHandler):
    protocol_version = "HTTP/1.1"
    disable_nagle_algorithm = True
    store: FakeKubeClient = None  # set by server factory

    def log_message(self, fmt, *args):  # quiet
        pass

    # -- helpers -------------------------------------------------------------
    def _send_raw(self, code: int, body: bytes) -> None:
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def _send(self, code: int, obj) -> None:
        body = json.dumps(obj).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def _body(self) -> dict:
        length = int(self.headers.get("Content-Length", 0))
        return json.loads(self.rfile.read(length)) if length else {}

    def _stream_watch(self, field_selector: str) -> None:
        """k8s `?watch=true`: chunked stream of newline-delimited watch
        events, until the client hangs up.  Only a `spec.nodeName=` selector
        is honored (what the informer asks for); phase transitions are the
        consumer's business, as with a real informer."""
        import queue

        sel = dict(
            kv.split("=", 1) for kv in field_selector.split(",") if "=" in kv
        )
        want_node = sel.get("spec.nodeName")
        sub = self.store.watch_subscribe()
        try:
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Transfer-Encoding", "chunked")
            self.end_headers()
            while True:
                try:
                    item = sub.get(timeout=5.0)
                    if item is None:       # server shutting down
                        break
                    node, line = item
                except queue.Empty:
                    # heartbeat chunk: detects a dead client, keeps NATs open
                    self.wfile.write(b"1\r\n\n\r\n")
                    self.wfile.flush()
                    continue
                if want_node and node != want_node:
                    continue
                payload = line + b"\n"
                self.wfile.write(
                    f"{len(payload):x}\r\n".encode() + payload + b"\r\n"
                )
                self.wfile.flush()
        except (BrokenPipeError, ConnectionResetError, OSError):
            pass
        finally:
            self.store.watch_unsubscribe(sub)
            self.close_connection = True

    def _route(self, method: str) -> None:
        try:
            self._route_inner(method)
        except KubeError as e:
            self._send(e.status, {"kind": "Status", "message": str(e)})
        except Exception as e:  # noqa: BLE001
            self._send(500, {"kind": "Status", "message": str(e)})

    def _route_inner(self, method: str) -> None:
        u = urlparse(self.path)
        parts = [p for p in u.path.split("/") if p]
        q = parse_qs(u.query)
        s = self.store

        if method == "GET" and parts == ["pods"]:
            # kubelet read-only view
            self._send_raw(200, s.kubelet_pods_raw())
            return
        if parts[:2] != ["api", "v1"]:
            self._send(404, {"message": "not found"})
            return
        rest = parts[2:]

        if rest == ["nodes"] and method == "GET":
            self._send(200, s.list_nodes())
            return
        if rest[:1] == ["nodes"] and len(rest) >= 2:
            name = rest[1]
            if method == "GET" and len(rest) == 2:
                self._send(200, s.get_node(name))
                return
            if method == "PATCH" and len(rest) == 2:
                self._send(200, s.patch_node(name, self._body()))
                return
            if method == "PATCH" and rest[2:] == ["status"]:
                self._send(200, s.patch_node_status(name, self._body()))
                return
        if rest == ["pods"] and method == "GET":
            if q.get("watch", ["false"])[0] == "true":
                self._stream_watch(q.get("fieldSelector", [""])[0])
                return
            self._send_raw(200, s.list_pods_raw(q.get("fieldSelector", [""])[0]))
            return
        if rest[:1] == ["namespaces"] and len(rest) == 3 and rest[2] == "events":
            if method == "POST":
                self._send(201, s.create_event(rest[1], self._body()))
                return
        if rest[:1] == ["namespaces"] and len(rest) >= 3 and rest[2] == "pods":
            ns = rest[1]
            if len(rest) == 3 and method == "POST":
                pod = self._body()
                pod.setdefault("metadata", {})["namespace"] = ns
                self._send(201, s.add_pod(pod))
                return
            if len(rest) == 3 and method == "GET":
                self._send_raw(200, s.list_pods_raw(namespace=ns))
                return
            if len(rest) == 4:
                name = rest[3]
                if method == "GET":
                    self._send_raw(200, s.get_pod_raw(ns, name))
                    return
                if method == "PATCH":
                    self._send(200, s.patch_pod(ns, name, self._body()))
                    return
                if method == "DELETE":
                    s.delete_pod(ns, name)
                    self._send(200, {"kind": "Status", "status": "Success"})
                    return
        self._send(404, {"message": f"no route {method} {u.path}"})

    def do_GET(self):
        self._route("GET")

    def do_POST(self):
        self._route("POST")

    def do_PATCH(self):
        self._route("PATCH")

    def do_DELETE(self):
        self._route("DELETE")


class FakeApiServer:
    """Threaded HTTP apiserver around a FakeKubeClient store."""

    def __init__(self, store: Optional[FakeKubeClient] = None, port: int = 0):
        self.store = store or FakeKubeClient()
        handler = type("BoundHandler", (_Handler,), {"store": self.store})
        self._httpd = TrackedThreadingHTTPServer(("127.0.0.1", port), handler)
        self.port = self._httpd.server_port
        self._thread = threading.Thread(
            target=self._httpd.serve_forever, name="fake-apiserver", daemon=True
        )

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.port}"

    def start(self) -> "FakeApiServer":
        self._thread.start()
        return self

    def stop(self) -> None:
        self.store.watch_close_all()
        self._httpd.shutdown()
        self._httpd.server_close()
        self._httpd.stop_all_connections()


def main(argv=None) -> int:
    """Standalone fake apiserver: `python -m gpushare_amd.cluster.fakeapiserver`.

    Used by bench.py so the apiserver is its own process (as in a real
    cluster) instead of sharing the plugin's GIL."""
    import argparse
    import sys
    import time

    p = argparse.ArgumentParser(prog="gpushare-fake-apiserver")
    p.add_argument("--port", type=int, default=0)
    p.add_argument("--node", default="bench-node")
    args = p.parse_args(argv)

    store = FakeKubeClient(node_name=args.node)
    server = FakeApiServer(store=store, port=args.port).start()
    print(f"READY {server.url}", flush=True)
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        server.stop()
    return 0


if __name__ == "__main__":
    import sys

    sys.exit(main())
