"""Fake kube-apiserver over real HTTP.

Wraps :class:`FakeKubeClient`'s in-memory store with the REST surface that
``RestKubeClient`` and the kubelet-pods path speak, so

- unit tests exercise the *real* HTTP client code, and
- the bench harness (bench.py) can run one shared apiserver that plugin,
  scheduler-extender, and N churn-generator ranks all talk to over
  127.0.0.1 — the same process topology as a real node.

Served by :class:`~gpushare_amd.cluster.fasthttp.FastHTTPServer` (minimal
keep-alive HTTP) so the harness apiserver is not the measured ceiling —
a real kube-apiserver is far faster than a Python stdlib HTTP stack.

Endpoints (subset of the k8s API the framework uses):
  GET    /api/v1/nodes/{name}
  PATCH  /api/v1/nodes/{name}            (metadata: topology annotation)
  PATCH  /api/v1/nodes/{name}/status
  GET    /api/v1/pods?fieldSelector=...[&watch=true]
  GET    /api/v1/namespaces/{ns}/pods/{name}
  PATCH  /api/v1/namespaces/{ns}/pods/{name}
  POST   /api/v1/namespaces/{ns}/pods          (bench: pod creation)
  DELETE /api/v1/namespaces/{ns}/pods/{name}   (bench: pod deletion)
  POST   /api/v1/namespaces/{ns}/events
  GET    /pods/                                 (kubelet read-only view)
"""

from __future__ import annotations

import json
import queue
from typing import Optional
from urllib.parse import parse_qs, urlparse

from .fasthttp import FastHTTPServer
from .kubeclient import FakeKubeClient, KubeError


def _json_bytes(obj) -> bytes:
    return json.dumps(obj).encode()


class FakeApiServer:
    """HTTP apiserver around a FakeKubeClient store (fasthttp-served)."""

    def __init__(self, store: Optional[FakeKubeClient] = None, port: int = 0):
        self.store = store or FakeKubeClient()
        self._httpd = FastHTTPServer(self._handle, port=port)
        self.port = self._httpd.port

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.port}"

    def start(self) -> "FakeApiServer":
        self._httpd.start()
        return self

    def stop(self) -> None:
        self.store.watch_close_all()
        self._httpd.stop()

    # ------------------------------------------------------------------ #
    def _handle(self, method: str, path: str, raw_body: bytes):
        try:
            return self._route(method, path, raw_body)
        except KubeError as e:
            return e.status, _json_bytes(
                {"kind": "Status", "message": str(e)}
            )
        except Exception as e:  # noqa: BLE001
            return 500, _json_bytes({"kind": "Status", "message": str(e)})

    def _route(self, method: str, path: str, raw_body: bytes):
        u = urlparse(path)
        parts = [p for p in u.path.split("/") if p]
        q = parse_qs(u.query)
        s = self.store
        body = lambda: json.loads(raw_body) if raw_body else {}  # noqa: E731

        if method == "GET" and parts == ["pods"]:
            return 200, s.kubelet_pods_raw()          # kubelet read-only view
        if parts[:2] != ["api", "v1"]:
            return 404, b'{"message":"not found"}'
        rest = parts[2:]

        if rest == ["nodes"] and method == "GET":
            return 200, _json_bytes(s.list_nodes())
        if rest[:1] == ["nodes"] and len(rest) >= 2:
            name = rest[1]
            if method == "GET" and len(rest) == 2:
                return 200, _json_bytes(s.get_node(name))
            if method == "PATCH" and len(rest) == 2:
                return 200, _json_bytes(s.patch_node(name, body()))
            if method == "PATCH" and rest[2:] == ["status"]:
                return 200, _json_bytes(s.patch_node_status(name, body()))
        if rest == ["pods"] and method == "GET":
            selector = q.get("fieldSelector", [""])[0]
            if q.get("watch", ["false"])[0] == "true":
                return "stream", self._watch_stream(selector)
            return 200, s.list_pods_raw(selector)
        if rest[:1] == ["namespaces"] and len(rest) == 3 and rest[2] == "events":
            if method == "POST":
                return 201, _json_bytes(s.create_event(rest[1], body()))
        if rest[:1] == ["namespaces"] and len(rest) >= 3 and rest[2] == "pods":
            ns = rest[1]
            if len(rest) == 3 and method == "POST":
                pod = body()
                pod.setdefault("metadata", {})["namespace"] = ns
                return 201, _json_bytes(s.add_pod(pod))
            if len(rest) == 3 and method == "GET":
                return 200, s.list_pods_raw(namespace=ns)
            if len(rest) == 4:
                name = rest[3]
                if method == "GET":
                    return 200, s.get_pod_raw(ns, name)
                if method == "PATCH":
                    return 200, _json_bytes(s.patch_pod(ns, name, body()))
                if method == "DELETE":
                    s.delete_pod(ns, name)
                    return 200, b'{"kind":"Status","status":"Success"}'
        return 404, _json_bytes({"message": f"no route {method} {u.path}"})

    # ------------------------------------------------------------------ #
    def _watch_stream(self, field_selector: str):
        """k8s `?watch=true`: chunked stream of newline-delimited watch
        events until the client hangs up or the store shuts down.  Only a
        `spec.nodeName=` selector is honored (what the informer asks for);
        phase transitions are the consumer's business, as with a real
        informer."""
        sel = dict(
            kv.split("=", 1) for kv in field_selector.split(",") if "=" in kv
        )
        want_node = sel.get("spec.nodeName")
        store = self.store

        def run(conn) -> None:
            sub = store.watch_subscribe()
            try:
                while True:
                    try:
                        item = sub.get(timeout=5.0)
                        if item is None:        # server shutting down
                            break
                        node, line = item
                    except queue.Empty:
                        # heartbeat chunk: detects a dead client
                        conn.sendall(b"1\r\n\n\r\n")
                        continue
                    if want_node and node != want_node:
                        continue
                    payload = line + b"\n"
                    conn.sendall(
                        f"{len(payload):x}\r\n".encode() + payload + b"\r\n"
                    )
            except (BrokenPipeError, ConnectionResetError, OSError):
                pass
            finally:
                store.watch_unsubscribe(sub)

        return run


def main(argv=None) -> int:
    """Standalone fake apiserver: `python -m gpushare_amd.cluster.fakeapiserver`.

    Used by bench.py so the apiserver is its own process (as in a real
    cluster) instead of sharing the plugin's GIL."""
    import argparse
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
