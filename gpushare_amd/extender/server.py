"""Scheduler-extender webhook server (k8s HTTPExtender protocol).

Speaks the scheduler's extender webhook JSON (``ExtenderArgs`` →
``ExtenderFilterResult``, ``ExtenderBindingArgs`` → ``ExtenderBindingResult``)
at the same URL prefix the gpushare scheduler-extender uses, plus a
``release`` hook (informer delete event) and a ``packing`` report.

  POST /gpushare-scheduler/filter
  POST /gpushare-scheduler/prioritize   (binpack node scoring)
  POST /gpushare-scheduler/bind
  POST /gpushare-scheduler/release
  GET  /gpushare-scheduler/packing
"""

from __future__ import annotations

import json

from ..cluster.fasthttp import FastHTTPServer
from .core import GPUShareExtender


class ExtenderServer:
    """Webhook server on the minimal fasthttp stack (the filter/bind RTT is
    inside the measured scheduling path)."""

    def __init__(self, extender: GPUShareExtender, port: int = 0):
        self.extender = extender
        self._httpd = FastHTTPServer(self._handle, port=port)
        self.port = self._httpd.port
        self._thread = None  # kept for API compat (daemon flag setters)

    @property
    def url(self) -> str:
        return f"http://127.0.0.1:{self.port}"

    def start(self) -> "ExtenderServer":
        self._httpd.start()
        return self

    def stop(self) -> None:
        self._httpd.stop()

    # ------------------------------------------------------------------ #
    def _handle(self, method: str, path: str, raw_body: bytes):
        try:
            return self._route(method, path, raw_body)
        except Exception as e:  # noqa: BLE001
            return 500, json.dumps({"Error": str(e)}).encode()

    def _route(self, method: str, path: str, raw_body: bytes):
        ext = self.extender
        if method == "GET":
            if path == "/gpushare-scheduler/packing":
                return 200, json.dumps(ext.packing()).encode()
            return 404, b'{"message":"not found"}'
        if method != "POST":
            return 404, b'{"message":"not found"}'
        body = json.loads(raw_body) if raw_body else {}
        if path == "/gpushare-scheduler/filter":
            pod = body.get("Pod") or {}
            names = body.get("NodeNames") or [
                n.get("metadata", {}).get("name")
                for n in (body.get("Nodes") or {}).get("Items", [])
            ]
            ok = ext.filter(pod, [n for n in names if n])
            return 200, json.dumps(
                {"NodeNames": ok, "FailedNodes": {}, "Error": ""}
            ).encode()
        if path == "/gpushare-scheduler/prioritize":
            pod = body.get("Pod") or {}
            names = body.get("NodeNames") or [
                n.get("metadata", {}).get("name")
                for n in (body.get("Nodes") or {}).get("Items", [])
            ]
            return 200, json.dumps(
                ext.prioritize(pod, [n for n in names if n])
            ).encode()
        if path == "/gpushare-scheduler/bind":
            ns = body.get("PodNamespace", "default")
            name = body.get("PodName", "")
            node = body.get("Node", "")
            pod = ext.cached_pod(ns, name)
            if pod is None:
                pod = ext.kube.get_pod(ns, name)
            idx = ext.assume(pod, node)
            if idx is None:
                return 200, json.dumps(
                    {"Error": f"no GPU on {node} fits pod {ns}/{name}"}
                ).encode()
            return 200, b'{"Error": ""}'
        if path == "/gpushare-scheduler/release":
            pod = body.get("Pod") or {}
            node = body.get("Node") or pod.get("spec", {}).get("nodeName", "")
            ext.release(pod, node)
            return 200, b'{"Error": ""}'
        return 404, b'{"message":"not found"}'


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

    def prioritize(self, pod: dict, node_names: list[str]) -> list[dict]:
        return self._post(
            "/gpushare-scheduler/prioritize",
            {"Pod": pod, "NodeNames": node_names},
        )

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
