"""Minimal Kubernetes REST client + in-memory fake.

Apiserver surface (reference uses client-go for the same core verbs,
pkg/gpu/nvidia/podmanager.go):
  - GET  node                              (isolation label check)
  - PATCH node                             (topology annotation)
  - PATCH node /status                     (aliyun.com/gpu-count)
  - LIST pods (fieldSelector, all ns)      (pending-pod fallback path)
  - WATCH pods (chunked stream)            (informer fast path)
  - PATCH pod (strategic merge)            (ASSIGNED=true handshake)
  - POST/DELETE pod, POST event            (bench harness / observability)

plus the kubelet read-only endpoint:
  - GET https://<node>:10250/pods/         (primary pending-pod path,
    reference: pkg/kubelet/client/client.go:119-134)

``FakeKubeClient`` implements the same surface in memory with
resourceVersion bumping and injectable 409 conflicts, giving the unit tests
what the reference never had (no fake clientset anywhere).
"""

from __future__ import annotations

import json
import os
import ssl
import threading
from typing import Optional
from urllib.parse import urlencode

import yaml

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class KubeError(RuntimeError):
    def __init__(self, status: int, message: str):
        super().__init__(f"HTTP {status}: {message}")
        self.status = status


class ConflictError(KubeError):
    """Optimistic-lock conflict (HTTP 409) on a patch."""

    def __init__(self, message: str = "conflict"):
        super().__init__(409, message)


# --------------------------------------------------------------------------- #
# Real clients (persistent keep-alive sessions, cluster/httpconn.py)
# --------------------------------------------------------------------------- #

class RestKubeClient:
    """Apiserver client: in-cluster service account, or a kubeconfig file
    (env ``KUBECONFIG``), mirroring kubeInit (podmanager.go:29-57)."""

    def __init__(
        self,
        base_url: Optional[str] = None,
        token: Optional[str] = None,
        verify=None,
        timeout: float = 10.0,
    ):
        from .httpconn import HttpSession

        if base_url is None:
            base_url, token, verify = self._auto_config()
        headers = {"Accept": "application/json"}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        self._client = HttpSession(
            base_url, headers=headers, verify=verify, timeout=timeout
        )

    @staticmethod
    def _auto_config():
        kubeconfig = os.environ.get("KUBECONFIG")
        if kubeconfig and os.path.exists(kubeconfig):
            cfg = yaml.safe_load(open(kubeconfig))
            ctx_name = cfg.get("current-context")
            ctx = next(
                c["context"] for c in cfg["contexts"] if c["name"] == ctx_name
            )
            cluster = next(
                c["cluster"]
                for c in cfg["clusters"]
                if c["name"] == ctx["cluster"]
            )
            user = next(
                u["user"] for u in cfg["users"] if u["name"] == ctx["user"]
            )
            # verified TLS is the default: a kubeconfig CA —
            # path or inline certificate-authority-data — pins the
            # apiserver cert; only an explicit insecure-skip-tls-verify
            # turns verification off; neither present ⇒ system trust store
            ca_data = cluster.get("certificate-authority-data")
            verify: object = cluster.get("certificate-authority", None)
            if verify is None and ca_data:
                import base64

                verify = ssl.create_default_context(
                    cadata=base64.b64decode(ca_data).decode()
                )
            if cluster.get("insecure-skip-tls-verify"):
                verify = False
            token = user.get("token")
            if not token and user.get("client-certificate"):
                if isinstance(verify, ssl.SSLContext):
                    sslctx = verify
                else:
                    sslctx = ssl.create_default_context(
                        cafile=verify if isinstance(verify, str) else None
                    )
                if verify is False:
                    sslctx.check_hostname = False
                    sslctx.verify_mode = ssl.CERT_NONE
                sslctx.load_cert_chain(
                    user["client-certificate"], user.get("client-key")
                )
                verify = sslctx
            return cluster["server"], token, verify
        # in-cluster
        host = os.environ.get("KUBERNETES_SERVICE_HOST", "kubernetes.default.svc")
        port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        token = None
        token_path = os.path.join(SA_DIR, "token")
        if os.path.exists(token_path):
            token = open(token_path).read().strip()
        ca = os.path.join(SA_DIR, "ca.crt")
        # no mounted CA ⇒ system trust store (verified) — never silently
        # insecure for apiserver traffic
        verify = ca if os.path.exists(ca) else None
        return f"https://{host}:{port}", token, verify

    # -- verbs ---------------------------------------------------------------
    _PATCH_HDRS = {"Content-Type": "application/strategic-merge-patch+json"}
    _JSON_HDRS = {"Content-Type": "application/json"}

    @staticmethod
    def _raise(status: int, body: bytes):
        text = body.decode(errors="replace")
        if status == 409:
            raise ConflictError(text)
        raise KubeError(status, text)

    def _check(self, status: int, body: bytes) -> dict:
        if status >= 400:
            self._raise(status, body)
        return json.loads(body)

    def get_node(self, name: str) -> dict:
        return self._check(*self._client.request("GET", f"/api/v1/nodes/{name}"))

    def list_nodes(self) -> dict:
        return self._check(*self._client.request("GET", "/api/v1/nodes"))

    def patch_node_status(self, name: str, patch: dict) -> dict:
        return self._check(
            *self._client.request(
                "PATCH",
                f"/api/v1/nodes/{name}/status",
                body=json.dumps(patch).encode(),
                headers=self._PATCH_HDRS,
            )
        )

    def patch_node(self, name: str, patch: dict) -> dict:
        """Strategic-merge patch of node metadata (topology annotation)."""
        return self._check(
            *self._client.request(
                "PATCH",
                f"/api/v1/nodes/{name}",
                body=json.dumps(patch).encode(),
                headers=self._PATCH_HDRS,
            )
        )

    def list_pods(self, field_selector: str = "", namespace: str = "") -> dict:
        path = (
            f"/api/v1/namespaces/{namespace}/pods" if namespace else "/api/v1/pods"
        )
        if field_selector:
            path += "?" + urlencode({"fieldSelector": field_selector})
        return self._check(*self._client.request("GET", path))

    def get_pod(self, namespace: str, name: str) -> dict:
        return self._check(
            *self._client.request(
                "GET", f"/api/v1/namespaces/{namespace}/pods/{name}"
            )
        )

    def create_pod(self, pod: dict, namespace: str = "default") -> None:
        status, body = self._client.request(
            "POST",
            f"/api/v1/namespaces/{namespace}/pods",
            body=json.dumps(pod).encode(),
            headers=self._JSON_HDRS,
        )
        if status >= 400:
            self._raise(status, body)

    def delete_pod(self, namespace: str, name: str) -> None:
        status, body = self._client.request(
            "DELETE", f"/api/v1/namespaces/{namespace}/pods/{name}"
        )
        if status >= 400:
            self._raise(status, body)

    def patch_pod(self, namespace: str, name: str, patch: dict,
                  parse: bool = True):
        status, body = self._client.request(
            "PATCH",
            f"/api/v1/namespaces/{namespace}/pods/{name}",
            body=json.dumps(patch).encode(),
            headers=self._PATCH_HDRS,
        )
        if status >= 400:
            self._raise(status, body)
        return json.loads(body) if parse else None

    def watch_pods_stream(self, field_selector: str = ""):
        """Open a k8s watch on pods (chunked stream); returns (conn, resp).
        The caller reads newline-delimited watch events from ``resp`` and
        closes ``conn`` when done (used by cluster.informer.PodInformer).

        ``timeoutSeconds=300`` asks the apiserver to close the watch
        cleanly at 5 min (the informer reconnects); the socket timeout sits
        just above so a dead server is still detected on a quiet node."""
        params = {"watch": "true", "timeoutSeconds": "300"}
        if field_selector:
            params["fieldSelector"] = field_selector
        conn, resp = self._client.stream(
            "GET", "/api/v1/pods?" + urlencode(params), timeout=330.0
        )
        if resp.status >= 400:
            body = resp.read()
            conn.close()
            self._raise(resp.status, body)
        return conn, resp

    def create_event(self, namespace: str, event: dict) -> dict:
        return self._check(
            *self._client.request(
                "POST",
                f"/api/v1/namespaces/{namespace}/events",
                body=json.dumps(event).encode(),
                headers=self._JSON_HDRS,
            )
        )

    def close(self) -> None:
        self._client.close()


class KubeletClient:
    """Read-only kubelet client (GET /pods), bearer-token HTTPS with TLS
    verification off — matching the reference's forced insecure transport
    (client.go:75-99; the kubelet's serving cert is rarely CA-signed)."""

    def __init__(
        self,
        address: str = "127.0.0.1",
        port: int = 10250,
        token: Optional[str] = None,
        timeout: float = 10.0,
        scheme: str = "https",
        client_cert: str = "",
        client_key: str = "",
    ):
        from .httpconn import HttpSession, make_ssl_context

        headers = {"Accept": "application/json"}
        if token is None:
            token_path = os.path.join(SA_DIR, "token")
            if os.path.exists(token_path):
                token = open(token_path).read().strip()
        if token:
            headers["Authorization"] = f"Bearer {token}"
        # mTLS option (reference: --client-cert/--client-key flags feed the
        # kubelet client, cmd/nvidia/main.go:28-53); server verification
        # stays off either way (client.go:75-99 forces insecure)
        verify = False
        if client_cert:
            ctx = make_ssl_context(False)
            ctx.load_cert_chain(client_cert, client_key or None)
            verify = ctx
        self._client = HttpSession(
            f"{scheme}://{address}:{port}",
            headers=headers,
            verify=verify,
            timeout=timeout,
        )

    def get_node_running_pods(self) -> dict:
        status, body = self._client.request("GET", "/pods/")
        if status >= 400:
            raise KubeError(status, body.decode(errors="replace"))
        return json.loads(body)

    def close(self) -> None:
        self._client.close()


# --------------------------------------------------------------------------- #
# Fakes (tests + bench harness)
# --------------------------------------------------------------------------- #

class FakeKubeClient:
    """In-memory apiserver: same verbs, resourceVersion bumping, injectable
    conflicts; doubles as the kubelet /pods source via ``as_kubelet()``."""

    def __init__(self, node_name: str = "node-a"):
        self.node_name = node_name
        self._lock = threading.RLock()
        self._rv = 0
        self.nodes: dict[str, dict] = {
            node_name: {
                "metadata": {"name": node_name, "labels": {}},
                "status": {"capacity": {}, "allocatable": {}},
            }
        }
        self.pods: dict[tuple, dict] = {}
        self._pod_bytes: dict[tuple, bytes] = {}   # key -> encoded JSON
        self.events: list[dict] = []
        self.fail_next_pod_patches = 0   # inject N consecutive 409s
        self.patch_count = 0
        self.list_count = 0
        self.watch_count = 0
        self._watchers: list = []        # SimpleQueue per open watch stream

    # -- watch feed (k8s `?watch=true` semantics for the fake) ---------------
    def watch_subscribe(self):
        import queue

        q = queue.SimpleQueue()
        with self._lock:
            self._watchers.append(q)
            self.watch_count += 1
        return q

    def watch_unsubscribe(self, q) -> None:
        with self._lock:
            if q in self._watchers:
                self._watchers.remove(q)

    def watch_close_all(self) -> None:
        """Terminate every open watch stream (server shutdown): a ``None``
        sentinel makes the handler close its connection."""
        with self._lock:
            for q in self._watchers:
                q.put(None)
            self._watchers.clear()

    def _publish(self, event_type: str, key: tuple) -> None:
        """Queue a watch event; caller holds self._lock.  The event carries
        the pod's encoded bytes (DELETED: the final state before removal)."""
        if not self._watchers:
            return
        node = self.pods[key].get("spec", {}).get("nodeName", "")
        line = (
            b'{"type":"' + event_type.encode() + b'","object":'
            + self._pod_bytes[key] + b"}"
        )
        for q in self._watchers:
            q.put((node, line))

    def _bump(self, obj: dict) -> None:
        self._rv += 1
        obj.setdefault("metadata", {})["resourceVersion"] = str(self._rv)

    def _reencode(self, key: tuple) -> None:
        self._pod_bytes[key] = json.dumps(self.pods[key]).encode()

    # -- test helpers --------------------------------------------------------
    def add_pod(self, pod: dict) -> dict:
        with self._lock:
            key = (pod["metadata"].get("namespace", "default"), pod["metadata"]["name"])
            pod["metadata"].setdefault("namespace", "default")
            pod["metadata"].setdefault("uid", f"uid-{key[0]}-{key[1]}")
            self._bump(pod)
            added = key not in self.pods
            self.pods[key] = pod
            self._reencode(key)
            self._publish("ADDED" if added else "MODIFIED", key)
            return pod

    def delete_pod(self, namespace: str, name: str) -> None:
        with self._lock:
            key = (namespace, name)
            if key in self._pod_bytes:
                self._publish("DELETED", key)
            self.pods.pop(key, None)
            self._pod_bytes.pop(key, None)

    # -- apiserver verbs -----------------------------------------------------
    def get_node(self, name: str) -> dict:
        with self._lock:
            if name not in self.nodes:
                raise KubeError(404, f"node {name} not found")
            return json.loads(json.dumps(self.nodes[name]))

    def list_nodes(self) -> dict:
        with self._lock:
            return {
                "kind": "NodeList",
                "items": [json.loads(json.dumps(n)) for n in self.nodes.values()],
            }

    def patch_node_status(self, name: str, patch: dict) -> dict:
        with self._lock:
            node = self.nodes[name]
            for sect in ("capacity", "allocatable"):
                node["status"].setdefault(sect, {}).update(
                    patch.get("status", {}).get(sect, {})
                )
            self._bump(node)
            return json.loads(json.dumps(node))

    def patch_node(self, name: str, patch: dict) -> dict:
        with self._lock:
            if name not in self.nodes:
                raise KubeError(404, f"node {name} not found")
            node = self.nodes[name]
            md = patch.get("metadata", {})
            if md.get("annotations"):
                node["metadata"].setdefault("annotations", {}).update(
                    md["annotations"]
                )
            if md.get("labels"):
                node["metadata"].setdefault("labels", {}).update(md["labels"])
            self._bump(node)
            return json.loads(json.dumps(node))

    def list_pods(self, field_selector: str = "", namespace: str = "") -> dict:
        with self._lock:
            self.list_count += 1
            sel = dict(
                kv.split("=", 1) for kv in field_selector.split(",") if "=" in kv
            )
            items = []
            for pod in self.pods.values():
                if namespace and pod["metadata"]["namespace"] != namespace:
                    continue
                if "spec.nodeName" in sel and (
                    pod.get("spec", {}).get("nodeName") != sel["spec.nodeName"]
                ):
                    continue
                if "status.phase" in sel and (
                    pod.get("status", {}).get("phase") != sel["status.phase"]
                ):
                    continue
                items.append(json.loads(json.dumps(pod)))
            return {"kind": "PodList", "items": items}

    def get_pod(self, namespace: str, name: str) -> dict:
        with self._lock:
            key = (namespace, name)
            if key not in self.pods:
                raise KubeError(404, f"pod {namespace}/{name} not found")
            return json.loads(json.dumps(self.pods[key]))

    def patch_pod(self, namespace: str, name: str, patch: dict,
                  parse: bool = True):
        with self._lock:
            self.patch_count += 1
            if self.fail_next_pod_patches > 0:
                self.fail_next_pod_patches -= 1
                raise ConflictError()
            key = (namespace, name)
            if key not in self.pods:
                raise KubeError(404, f"pod {namespace}/{name} not found")
            pod = self.pods[key]
            anns = patch.get("metadata", {}).get("annotations")
            if anns:
                pod["metadata"].setdefault("annotations", {}).update(anns)
            self._bump(pod)
            self._reencode(key)
            self._publish("MODIFIED", key)
            return json.loads(json.dumps(pod)) if parse else None

    # -- raw-bytes accessors (fakeapiserver hot path: the per-pod JSON is
    # encoded once per mutation, a LIST response is a join of cached bytes) --
    def list_pods_raw(self, field_selector: str = "", namespace: str = "") -> bytes:
        sel = dict(
            kv.split("=", 1) for kv in field_selector.split(",") if "=" in kv
        )
        want_node = sel.get("spec.nodeName")
        want_phase = sel.get("status.phase")
        with self._lock:
            self.list_count += 1
            chunks = []
            for key, pod in self.pods.items():
                if namespace and key[0] != namespace:
                    continue
                if want_node and pod.get("spec", {}).get("nodeName") != want_node:
                    continue
                if want_phase and pod.get("status", {}).get("phase") != want_phase:
                    continue
                chunks.append(self._pod_bytes[key])
            return b'{"kind":"PodList","items":[' + b",".join(chunks) + b"]}"

    def kubelet_pods_raw(self) -> bytes:
        return self.list_pods_raw(field_selector=f"spec.nodeName={self.node_name}")

    def get_pod_raw(self, namespace: str, name: str) -> bytes:
        with self._lock:
            key = (namespace, name)
            if key not in self.pods:
                raise KubeError(404, f"pod {namespace}/{name} not found")
            return self._pod_bytes[key]

    def create_event(self, namespace: str, event: dict) -> dict:
        with self._lock:
            event.setdefault("metadata", {})["namespace"] = namespace
            self.events.append(event)
            return json.loads(json.dumps(event))

    def close(self) -> None:
        pass

    # -- kubelet view --------------------------------------------------------
    def as_kubelet(self) -> "FakeKubeletClient":
        return FakeKubeletClient(self)


class FakeKubeletClient:
    def __init__(self, store: FakeKubeClient, fail_times: int = 0):
        self._store = store
        self.fail_times = fail_times
        self.query_count = 0

    def get_node_running_pods(self) -> dict:
        self.query_count += 1
        if self.fail_times > 0:
            self.fail_times -= 1
            raise KubeError(500, "kubelet unavailable (injected)")
        with self._store._lock:
            items = [
                json.loads(json.dumps(p))
                for p in self._store.pods.values()
                if p.get("spec", {}).get("nodeName") == self._store.node_name
            ]
        return {"kind": "PodList", "items": items}

    def close(self) -> None:
        pass
