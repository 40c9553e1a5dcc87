"""Scheduler-extender webhook server (k8s HTTPExtender protocol).

Speaks the scheduler's extender webhook JSON (``ExtenderArgs`` →
``ExtenderFilterResult``, ``ExtenderBindingArgs`` → ``ExtenderBindingResult``)
at the same URL prefix the gpushare scheduler-extender uses, plus a
``release`` hook (informer delete event) and a ``packing`` report.

  POST /gpushare-scheduler/filter
  POST /gpushare-scheduler/bind
  POST /gpushare-scheduler/release
  GET  /gpushare-scheduler/packing
"""

from __future__ import annotations

import json
import threading
from http.server import BaseHTTPRequestHandler

from ..cluster.httpconn import TrackedThreadingHTTPServer
from .core import GPUShareExtender


class _Handler(BaseHTTPRequestHandler):
    protocol_version = "HTTP/1.1"
    disable_nagle_algorithm = True
    extender: GPUShareExtender = None  # bound by server factory

    def log_message(self, fmt, *args):
        pass

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

    def do_GET(self):
        if self.path == "/gpushare-scheduler/packing":
            self._send(200, self.extender.packing())
        else:
            self._send(404, {"message": "not found"})

    def do_POST(self):
        try:
            body = self._body()
            if self.path == "/gpushare-scheduler/filter":
                pod = body.get("Pod") or {}
                names = body.get("NodeNames") or [
                    n.get("metadata", {}).get("name")
                    for n in (body.get("Nodes") or {}).get("Items", [])
                ]
                ok = self.extender.filter(pod, [n for n in names if n])
                self._send(
                    200, {"NodeNames": ok, "FailedNodes": {}, "Error": ""}
                )
            elif self.path == "/gpushare-scheduler/bind":
                ns = body.get("PodNamespace", "default")
                name = body.get("PodName", "")
                node = body.get("Node", "")
                pod = self.extender.cached_pod(ns, name)
                if pod is None:
                    pod = self.extender.kube.get_pod(ns, name)
                idx = self.extender.assume(pod, node)
                if idx is None:
                    self._send(
                        200,
                        {"Error": f"no GPU on {node} fits pod {ns}/{name}"},
                    )
                else:
                    self._send(200, {"Error": ""})
            elif self.path == "/gpushare-scheduler/release":
                pod = body.get("Pod") or {}
                node = body.get("Node") or pod.get("spec", {}).get("nodeName", "")
                self.extender.release(pod, node)
                self._send(200, {"Error": ""})
            else:
                self._send(404, {"message": "not found"})
        except Exception as e:  # noqa: BLE001
            self._send(500, {"Error": str(e)})


class ExtenderServer:
    def __init__(self, extender: GPUShareExtender, port: int = 0):
        handler = type("BoundHandler", (_Handler,), {"extender": extender})
        self._httpd = TrackedThreadingHTTPServer(("127.0.0.1", port), handler)
        self.port = self._httpd.server_port
        self._thread = threading.Thread(
            target=self._httpd.serve_forever, name="extender", daemon=True
        )

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.port}"

    def start(self) -> "ExtenderServer":
        self._thread.start()
        return self

    def stop(self) -> None:
        self._httpd.shutdown()
        self._httpd.server_close()
        self._httpd.stop_all_connections()


class ExtenderClient:
    """Client used by churn generators (and by a scheduler integration)."""

    def __init__(self, url: str, timeout: float = 10.0):
        from ..cluster.httpconn import HttpSession

        self._client = HttpSession(url, timeout=timeout)

    def _post(self, path: str, obj: dict) -> dict:
        status, body = self._client.request(
            "POST", path, body=json.dumps(obj).encode(),
            headers={"Content-Type": "application/json"},
        )
        if status >= 400:
            raise RuntimeError(
                f"extender {path}: HTTP {status}: {body.decode(errors='replace')}"
            )
        return json.loads(body)

    def filter(self, pod: dict, node_names: list[str]) -> list[str]:
        return self._post(
            "/gpushare-scheduler/filter",
            {"Pod": pod, "NodeNames": node_names},
        )["NodeNames"]

    def bind(self, namespace: str, name: str, node: str) -> str:
        return self._post(
            "/gpushare-scheduler/bind",
            {"PodNamespace": namespace, "PodName": name, "Node": node},
        ).get("Error", "")

    def release(self, pod: dict, node: str = "") -> None:
        self._post("/gpushare-scheduler/release", {"Pod": pod, "Node": node})

    def packing(self) -> dict:
        status, body = self._client.request("GET", "/gpushare-scheduler/packing")
        if status >= 400:
            raise RuntimeError(f"extender packing: HTTP {status}")
        return json.loads(body)

    def close(self) -> None:
        self._client.close()
