"""PodManager: listing paths, retries, dedup, FIFO sort, cache, patches."""

import pytest

from gpushare_amd import consts
from gpushare_amd.cluster.kubeclient import FakeKubeClient, FakeKubeletClient
from gpushare_amd.cluster.podmanager import PodManager

from helpers import make_pod, plain_pod


@pytest.fixture
def kube():
    return FakeKubeClient(node_name="node-a")


def _pm(kube, **kw):
    kw.setdefault("kubelet_client", kube.as_kubelet())
    kw.setdefault("cache_ttl", 0.0)  # most tests want fresh lists
    kw.setdefault("kubelet_retries", 0)
    kw.setdefault("kubelet_retry_interval", 0.0)
    kw.setdefault("apiserver_retries", 0)
    kw.setdefault("apiserver_retry_interval", 0.0)
    return PodManager(kube, "node-a", **kw)


def test_pending_pods_filters_node_and_phase(kube):
    kube.add_pod(make_pod("on-node", 2))
    kube.add_pod(make_pod("other-node", 2, node="node-b"))
    kube.add_pod(plain_pod("running", phase="Running"))
    pm = _pm(kube)
    pods = pm.get_pending_pods()
    assert [p["metadata"]["name"] for p in pods] == ["on-node"]


def test_kubelet_fallback_to_apiserver(kube):
    kube.add_pod(make_pod("p1", 2))
    kubelet = FakeKubeletClient(kube, fail_times=100)
    pm = _pm(kube, kubelet_client=kubelet)
    pods = pm.get_pending_pods()
    assert len(pods) == 1
    assert kube.list_count >= 1  # fell back to apiserver LIST


def test_kubelet_retry_then_success(kube):
    kube.add_pod(make_pod("p1", 2))
    kubelet = FakeKubeletClient(kube, fail_times=2)
    pm = _pm(kube, kubelet_client=kubelet, kubelet_retries=3)
    pods = pm.get_pending_pods()
    assert len(pods) == 1
    assert kubelet.query_count == 3  # 2 failures + 1 success
    assert kube.list_count == 0      # never hit apiserver


def test_candidate_sort_fifo(kube):
    kube.add_pod(make_pod("new", 2, assume_time_ns=2000))
    kube.add_pod(make_pod("old", 2, assume_time_ns=1000))
    kube.add_pod(make_pod("assigned", 2, assigned="true"))
    pm = _pm(kube)
    names = [p["metadata"]["name"] for p in pm.get_candidate_pods()]
    assert names == ["old", "new"]


def test_cache_serves_within_ttl(kube):
    kube.add_pod(make_pod("p1", 2))
    kubelet = kube.as_kubelet()
    pm = _pm(kube, kubelet_client=kubelet, cache_ttl=60.0)
    pm.get_pending_pods()
    pm.get_pending_pods()
    pm.get_pending_pods()
    assert kubelet.query_count == 1  # served from cache
    pm.get_pending_pods(force_refresh=True)
    assert kubelet.query_count == 2


def test_mark_assigned_conflict_retry(kube):
    pod = kube.add_pod(make_pod("p1", 2))
    kube.fail_next_pod_patches = 1
    pm = _pm(kube)
    assert pm.mark_assigned(pod) is True
    assert kube.patch_count == 2
    stored = kube.get_pod("default", "p1")
    assert stored["metadata"]["annotations"][consts.ENV_ASSIGNED_FLAG] == "true"


def test_mark_assigned_gives_up_after_retries(kube):
    pod = kube.add_pod(make_pod("p1", 2))
    kube.fail_next_pod_patches = 5
    pm = _pm(kube)
    assert pm.mark_assigned(pod, retries=1) is False


def test_patch_gpu_count(kube):
    pm = _pm(kube)
    pm.patch_gpu_count(8)
    node = kube.get_node("node-a")
    assert node["status"]["capacity"][consts.RESOURCE_COUNT] == "8"
    assert node["status"]["allocatable"][consts.RESOURCE_COUNT] == "8"
    # idempotent second patch short-circuits
    before = kube._rv
    pm.patch_gpu_count(8)
    assert kube._rv == before


def test_isolation_disabled_label(kube):
    pm = _pm(kube)
    assert pm.isolation_disabled() is False
    kube.nodes["node-a"]["metadata"]["labels"][consts.LABEL_DISABLE_ISOLATION] = "true"
    assert pm.isolation_disabled() is True


def test_dedup_by_uid(kube):
    pod = make_pod("dup", 2)
    kube.add_pod(pod)
    # simulate kubelet returning the same pod twice
    class DupKubelet:
        def get_node_running_pods(self):
            import json
            p = json.loads(json.dumps(pod))
            return {"items": [p, p]}

    pm = _pm(kube, kubelet_client=DupKubelet())
    assert len(pm.get_pending_pods()) == 1


def test_real_kubelet_client_over_http():
    """KubeletClient (the real HTTP client, reference client.go:119-134)
    against the fake apiserver's kubelet read-only /pods view."""
    from gpushare_amd.cluster.fakeapiserver import FakeApiServer
    from gpushare_amd.cluster.kubeclient import FakeKubeClient, KubeletClient

    from helpers import make_pod

    server = FakeApiServer(store=FakeKubeClient(node_name="node-a")).start()
    try:
        server.store.add_pod(make_pod("p1", node="node-a", mem=4))
        server.store.add_pod(make_pod("elsewhere", node="node-b", mem=4))
        kc = KubeletClient(
            address="127.0.0.1", port=server.port, scheme="http", token="t"
        )
        pods = kc.get_node_running_pods()
        names = [p["metadata"]["name"] for p in pods["items"]]
        assert names == ["p1"]  # node-scoped view
        kc.close()
    finally:
        server.stop()


def test_podgetter_cli_over_tls(tmp_path, capsys):
    """gpushare-podgetter against a TLS kubelet endpoint (the production
    shape: self-signed serving cert, bearer token, verification off)."""
    import json
    import ssl
    import subprocess

    from gpushare_amd.cli import podgetter
    from gpushare_amd.cluster.fasthttp import FastHTTPServer

    cert, key = tmp_path / "t.crt", tmp_path / "t.key"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-days", "1", "-subj", "/CN=127.0.0.1",
         "-keyout", str(key), "-out", str(cert)],
        check=True, capture_output=True,
    )
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
    ctx.load_cert_chain(str(cert), str(key))
    podlist = {"kind": "PodList", "items": []}

    srv = FastHTTPServer(
        lambda m, p, b: (200, json.dumps(podlist).encode()), ssl_context=ctx
    ).start()
    try:
        rc = podgetter.main(
            ["--kubelet-port", str(srv.port), "--token", "tok"]
        )
        assert rc == 0
        assert json.loads(capsys.readouterr().out) == podlist
    finally:
        srv.stop()
