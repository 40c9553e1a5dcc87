"""Metrics + k8s Event emission."""

import pytest

from gpushare_amd import metrics
from gpushare_amd.allocator import Allocator
from gpushare_amd.cluster.events import EventRecorder, NullRecorder
from gpushare_amd.cluster.kubeclient import FakeKubeClient
from gpushare_amd.cluster.podmanager import PodManager
from gpushare_amd.device.mock_source import MockSource

from helpers import make_pod
from test_allocator import _request


def test_event_recorder_emits_node_event():
    kube = FakeKubeClient("node-a")
    rec = EventRecorder(kube, "node-a")
    assert rec.emit("GPUUnhealthy", "GPU 1: thermal throttle") is True
    assert len(kube.events) == 1
    ev = kube.events[0]
    assert ev["involvedObject"] == {
        "kind": "Node",
        "name": "node-a",
        "uid": "node-a",
    }
    assert ev["reason"] == "GPUUnhealthy"
    assert ev["type"] == "Warning"


def test_event_recorder_never_raises():
    class BrokenKube:
        def create_event(self, ns, ev):
            raise RuntimeError("apiserver down")

    rec = EventRecorder(BrokenKube(), "node-a")
    assert rec.emit("X", "y") is False
    assert NullRecorder().emit("X", "y") is False


def test_allocator_emits_event_on_poisoned():
    kube = FakeKubeClient("node-a")
    pm = PodManager(kube, "node-a", kubelet_client=kube.as_kubelet(),
                    cache_ttl=0.0, kubelet_retries=0,
                    kubelet_retry_interval=0.0, apiserver_retries=0)
    gpus = MockSource.from_spec("8x288GiB").devices()
    alloc = Allocator(gpus, pm, event_recorder=EventRecorder(kube, "node-a"))
    alloc.allocate(_request([9]))
    assert any(e["reason"] == "GPUShareAllocateFailed" for e in kube.events)


def test_events_over_http_apiserver():
    from gpushare_amd.cluster.fakeapiserver import FakeApiServer
    from gpushare_amd.cluster.kubeclient import RestKubeClient

    api = FakeApiServer().start()
    try:
        client = RestKubeClient(base_url=api.url)
        rec = EventRecorder(client, "node-a")
        assert rec.emit("TestReason", "hello") is True
        assert api.store.events[0]["message"] == "hello"
        client.close()
    finally:
        api.stop()


@pytest.mark.skipif(not metrics.AVAILABLE, reason="prometheus_client missing")
def test_metrics_observed_through_allocate():
    before = metrics.ALLOCATE_TOTAL.labels("ok")._value.get()
    kube = FakeKubeClient("node-a")
    pm = PodManager(kube, "node-a", kubelet_client=kube.as_kubelet(),
                    cache_ttl=0.0, kubelet_retries=0,
                    kubelet_retry_interval=0.0, apiserver_retries=0)
    gpus = MockSource.from_spec("8x288GiB").devices()
    alloc = Allocator(gpus, pm)
    kube.add_pod(make_pod("m1", 4, gpu_idx=0))
    alloc.allocate(_request([4]))
    assert metrics.ALLOCATE_TOTAL.labels("ok")._value.get() == before + 1


@pytest.mark.skipif(not metrics.AVAILABLE, reason="prometheus_client missing")
def test_metrics_http_endpoint():
    from gpushare_amd.cluster.httpconn import HttpSession

    server = metrics.serve(0)
    try:
        port = server.server_port
        metrics.observe_informer_state(True)
        metrics.observe_vram_usage({0: 123})
        s = HttpSession(f"http://127.0.0.1:{port}")
        status, raw = s.request("GET", "/metrics")
        body = raw.decode()
        s.close()
        assert status == 200
        assert "gpushare_allocate_total" in body
        assert "gpushare_fake_devices" in body
        assert "gpushare_informer_synced 1.0" in body
        assert 'gpushare_vram_used_bytes{gpu="0"} 123.0' in body
    finally:
        server.shutdown()
