"""Fault injection: control-plane outages must degrade, not break.

SURVEY §5.3: the reference has retries but no fault-injection tests at
all.  These cover the plugin's behavior when the apiserver disappears
mid-operation and when it returns: Allocate answers (poisoned envs, never
a hung/failed RPC), the informer reconnects, and service resumes."""

from __future__ import annotations

import time

import pytest

from gpushare_amd import consts
from gpushare_amd.allocator import Allocator
from gpushare_amd.cluster.fakeapiserver import FakeApiServer
from gpushare_amd.cluster.informer import PodInformer
from gpushare_amd.cluster.kubeclient import FakeKubeClient, RestKubeClient
from gpushare_amd.cluster.podmanager import PodManager
from gpushare_amd.device.mock_source import MockSource
from gpushare_amd.deviceplugin import v1beta1 as api_pb

from helpers import make_pod

NODE = "node-a"


def _request(units, uuid="mock-00"):
    req = api_pb.AllocateRequest()
    cr = req.container_requests.add()
    for n in range(units):
        cr.devicesIDs.append(f"{uuid}-_-{n}")
    return req


def wait_for(pred, timeout=10.0, interval=0.01):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(interval)
    return False


@pytest.fixture()
def stack():
    server = FakeApiServer(store=FakeKubeClient(node_name=NODE)).start()
    kube = RestKubeClient(base_url=server.url, timeout=2.0)
    informer = PodInformer(
        RestKubeClient(base_url=server.url, timeout=2.0),
        NODE,
        reconnect_backoff=0.05,
    ).start()
    assert informer.wait_synced(5)
    pm = PodManager(
        kube,
        NODE,
        kubelet_client=None,
        query_kubelet=False,
        informer=informer,
        apiserver_retries=0,
        cache_ttl=0.0,
    )
    gpus = MockSource.from_spec("2x16GiB").devices()
    alloc = Allocator(gpus, pm)
    yield server, informer, alloc
    informer.stop()
    server.stop()


def test_apiserver_outage_and_recovery(stack):
    server, informer, alloc = stack
    port = server.port

    # healthy allocate
    server.store.add_pod(make_pod("p1", node=NODE, mem=8, gpu_idx=0))
    assert wait_for(lambda: len(informer.pods()) == 1)
    resp = alloc.allocate(_request(8))
    assert resp.container_responses[0].envs[consts.ENV_RESOURCE_INDEX] == "0"

    # apiserver dies
    server.stop()
    assert wait_for(lambda: not informer.synced, timeout=10)

    # allocate during the outage: bounded, poisoned, never raises
    t0 = time.monotonic()
    resp = alloc.allocate(_request(8))
    elapsed = time.monotonic() - t0
    envs = resp.container_responses[0].envs
    assert envs[consts.ENV_RESOURCE_INDEX] == "-1"
    assert "no-gpu-has" in envs[consts.ENV_ROCR_VISIBLE]
    assert elapsed < 10.0, f"outage allocate took {elapsed:.1f}s"

    # apiserver returns on the same port with a fresh assumed pod
    store = FakeKubeClient(node_name=NODE)
    store.add_pod(make_pod("p2", node=NODE, mem=8, gpu_idx=1))
    server2 = FakeApiServer(store=store, port=port).start()
    try:
        assert informer.wait_synced(15)
        assert wait_for(lambda: len(informer.pods()) == 1)
        resp = alloc.allocate(_request(8))
        envs = resp.container_responses[0].envs
        assert envs[consts.ENV_RESOURCE_INDEX] == "1"
        anns = store.get_pod("default", "p2")["metadata"]["annotations"]
        assert anns[consts.ENV_ASSIGNED_FLAG] == "true"
    finally:
        server2.stop()


def test_patch_conflict_storm(stack):
    """Every ASSIGNED patch 409s: the allocate must give up cleanly with a
    poisoned response and unclaim the pod (reference allows one retry,
    allocate.go:138-144)."""
    server, informer, alloc = stack
    server.store.add_pod(make_pod("p1", node=NODE, mem=8, gpu_idx=0))
    assert wait_for(lambda: len(informer.pods()) == 1)
    server.store.fail_next_pod_patches = 99
    resp = alloc.allocate(_request(8))
    assert resp.container_responses[0].envs[consts.ENV_RESOURCE_INDEX] == "-1"
    server.store.fail_next_pod_patches = 0
    # pod is unclaimed: the retried allocate (kubelet retries the pod
    # lifecycle) succeeds
    resp = alloc.allocate(_request(8))
    assert resp.container_responses[0].envs[consts.ENV_RESOURCE_INDEX] == "0"


def test_concurrent_allocates_informer_mode(stack):
    """16 same-size pods, 16 concurrent Allocates against the watch-backed
    store (the production informer path): every pod assigned exactly once,
    no poisoned responses, no double assignment."""
    import threading

    server, informer, alloc = stack
    for i in range(16):
        server.store.add_pod(
            make_pod(f"p{i}", node=NODE, mem=4, gpu_idx=i % 2, assume_time_ns=i)
        )
    assert wait_for(lambda: len(informer.pods()) == 16)
    results = []
    lock = threading.Lock()

    def run():
        resp = alloc.allocate(_request(4))
        with lock:
            results.append(
                resp.container_responses[0].envs[consts.ENV_RESOURCE_INDEX]
            )

    threads = [threading.Thread(target=run) for _ in range(16)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert "-1" not in results
    assigned = [
        p
        for p in server.store.pods.values()
        if p["metadata"]["annotations"][consts.ENV_ASSIGNED_FLAG] == "true"
    ]
    assert len(assigned) == 16
