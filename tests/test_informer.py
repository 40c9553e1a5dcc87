"""PodInformer: LIST+WATCH cache over the fake apiserver's watch endpoint.

The informer replaces the reference's per-Allocate remote LIST
(allocate.go:59-62 + podmanager.go:125-160) with a push-updated store;
these tests cover event propagation, reconnect, resourceVersion
reconciliation, and the PodManager/Allocator integration (informer fast
path + authoritative fallback).
"""

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


def _request(container_units, uuid="mock-00"):
    req = api_pb.AllocateRequest()
    n = 0
    for units in container_units:
        cr = req.container_requests.add()
        for _ in range(units):
            cr.devicesIDs.append(f"{uuid}-_-{n}")
            n += 1
    return req


@pytest.fixture()
def api():
    server = FakeApiServer(store=FakeKubeClient(node_name=NODE)).start()
    yield server
    server.stop()


def wait_for(pred, timeout=5.0, interval=0.005):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(interval)
    return False


def make_informer(api, **kw):
    return PodInformer(RestKubeClient(base_url=api.url), NODE, **kw)


class TestPodInformer:
    def test_sync_sees_preexisting_pods(self, api):
        api.store.add_pod(make_pod("p1", node=NODE, mem=4))
        api.store.add_pod(make_pod("other", node="node-b", mem=4))
        inf = make_informer(api).start()
        try:
            assert inf.wait_synced(5)
            pods = inf.pods()
            assert [p["metadata"]["name"] for p in pods] == ["p1"]
        finally:
            inf.stop()

    def test_add_patch_delete_propagate(self, api):
        inf = make_informer(api).start()
        assert inf.wait_synced(5)
        try:
            api.store.add_pod(make_pod("p1", node=NODE, mem=8))
            assert wait_for(lambda: len(inf.pods()) == 1)

            api.store.patch_pod(
                "default", "p1",
                {"metadata": {"annotations": {"x": "y"}}},
            )
            assert wait_for(
                lambda: any(
                    p["metadata"].get("annotations", {}).get("x") == "y"
                    for p in inf.pods()
                )
            )

            api.store.delete_pod("default", "p1")
            assert wait_for(lambda: len(inf.pods()) == 0)
        finally:
            inf.stop()

    def test_other_node_events_filtered(self, api):
        inf = make_informer(api).start()
        assert inf.wait_synced(5)
        try:
            api.store.add_pod(make_pod("elsewhere", node="node-b", mem=8))
            api.store.add_pod(make_pod("here", node=NODE, mem=8))
            assert wait_for(lambda: len(inf.pods()) == 1)
            time.sleep(0.05)
            assert [p["metadata"]["name"] for p in inf.pods()] == ["here"]
        finally:
            inf.stop()

    def test_reconnect_after_server_restart(self, api):
        inf = make_informer(api, reconnect_backoff=0.05).start()
        assert inf.wait_synced(5)
        port = api.port
        api.stop()
        assert wait_for(lambda: not inf.synced, timeout=15)
        # new server on the same port with existing state
        store = FakeKubeClient(node_name=NODE)
        store.add_pod(make_pod("survivor", node=NODE, mem=8))
        server2 = FakeApiServer(store=store, port=port).start()
        try:
            assert inf.wait_synced(15)
            assert wait_for(
                lambda: [p["metadata"]["name"] for p in inf.pods()]
                == ["survivor"]
            )
            assert inf.reconnects >= 1
        finally:
            inf.stop()
            server2.stop()

    def test_stale_event_does_not_resurrect(self, api):
        """A DELETED event with an older resourceVersion than the stored
        object must not clobber a newer ADDED (rv reconciliation)."""
        inf = make_informer(api).start()
        assert inf.wait_synced(5)
        try:
            api.store.add_pod(make_pod("p1", node=NODE, mem=8))
            assert wait_for(lambda: len(inf.pods()) == 1)
            newer = dict(inf.pods()[0])
            stale = {
                "metadata": {
                    **newer["metadata"],
                    "resourceVersion": "0",
                }
            }
            inf._apply("DELETED", stale)
            assert len(inf.pods()) == 1  # stale delete ignored
        finally:
            inf.stop()


class TestPodManagerInformerPath:
    def test_informer_serves_pending_without_remote_list(self, api):
        inf = make_informer(api).start()
        assert inf.wait_synced(5)
        pm = PodManager(
            RestKubeClient(base_url=api.url),
            NODE,
            kubelet_client=None,
            query_kubelet=False,
            informer=inf,
            apiserver_retries=0,
        )
        try:
            api.store.add_pod(make_pod("p1", node=NODE, mem=8, gpu_idx=0))
            assert wait_for(lambda: len(pm.get_candidate_pods()) == 1)
            before = api.store.list_count
            for _ in range(20):
                assert len(pm.get_candidate_pods()) == 1
            assert api.store.list_count == before  # zero remote lists
        finally:
            inf.stop()

    def test_authoritative_bypasses_informer(self, api):
        inf = make_informer(api).start()
        assert inf.wait_synced(5)
        pm = PodManager(
            RestKubeClient(base_url=api.url),
            NODE,
            kubelet_client=None,
            query_kubelet=False,
            informer=inf,
            apiserver_retries=0,
        )
        try:
            before = api.store.list_count
            pm.get_candidate_pods(force_refresh=True, authoritative=True)
            assert api.store.list_count == before + 1
        finally:
            inf.stop()

    def test_allocate_via_informer(self, api):
        """Full allocate against the informer-backed PodManager: the match
        must come from the watch store (no remote LIST), the patch must
        land on the apiserver."""
        inf = make_informer(api).start()
        assert inf.wait_synced(5)
        gpus = MockSource.from_spec("2x16GiB").devices()
        pm = PodManager(
            RestKubeClient(base_url=api.url),
            NODE,
            kubelet_client=None,
            query_kubelet=False,
            informer=inf,
            apiserver_retries=0,
        )
        alloc = Allocator(gpus, pm)
        try:
            api.store.add_pod(make_pod("p1", node=NODE, mem=8, gpu_idx=1))
            assert wait_for(lambda: len(pm.get_candidate_pods()) == 1)
            lists_before = api.store.list_count

            resp = alloc.allocate(_request([8]))
            envs = resp.container_responses[0].envs
            assert envs[consts.ENV_RESOURCE_INDEX] == "1"
            assert api.store.list_count == lists_before
            pod = api.store.get_pod("default", "p1")
            anns = pod["metadata"]["annotations"]
            assert anns[consts.ENV_ASSIGNED_FLAG] == "true"
        finally:
            inf.stop()


class TestDegradedMode:
    def test_allocate_works_with_unsynced_informer(self, api):
        """Watch endpoint unreachable (informer never syncs) ⇒ the
        allocator transparently uses the remote-list path — the informer
        is an accelerator, not a correctness dependency."""

        class NeverSyncs:
            synced = False
            version = 0

            def pending_pods(self):  # pragma: no cover
                raise AssertionError("must not be consulted while unsynced")

        gpus = MockSource.from_spec("2x16GiB").devices()
        pm = PodManager(
            RestKubeClient(base_url=api.url),
            NODE,
            kubelet_client=None,
            query_kubelet=False,
            informer=NeverSyncs(),
            cache_ttl=0.0,
            apiserver_retries=0,
        )
        alloc = Allocator(gpus, pm)
        api.store.add_pod(make_pod("p1", node=NODE, mem=8, gpu_idx=1))
        resp = alloc.allocate(_request([8]))
        envs = resp.container_responses[0].envs
        assert envs[consts.ENV_RESOURCE_INDEX] == "1"


class TestInformerModelCheck:
    def test_random_op_sequences_converge_to_server_state(self, api):
        """Model-based check: after any seeded sequence of creates,
        patches and deletes, the informer store converges to exactly the
        server's node-scoped pod set."""
        import random

        inf = make_informer(api).start()
        assert inf.wait_synced(5)
        try:
            rng = random.Random(1234)
            live: set[str] = set()
            counter = 0
            for _ in range(300):
                op = rng.random()
                if op < 0.45 or not live:
                    name = f"p{counter}"
                    counter += 1
                    node = NODE if rng.random() < 0.8 else "node-b"
                    api.store.add_pod(make_pod(name, node=node, mem=4))
                    if node == NODE:
                        live.add(name)
                elif op < 0.75:
                    name = rng.choice(sorted(live))
                    api.store.patch_pod(
                        "default", name,
                        {"metadata": {"annotations": {"seq": str(counter)}}},
                    )
                    counter += 1
                else:
                    name = rng.choice(sorted(live))
                    api.store.delete_pod("default", name)
                    live.discard(name)
            assert wait_for(
                lambda: {
                    p["metadata"]["name"] for p in inf.pods()
                } == live,
                timeout=10,
            ), (
                f"store diverged: informer="
                f"{sorted(p['metadata']['name'] for p in inf.pods())[:10]}... "
                f"expected {len(live)} pods"
            )
            # annotations converge too (latest resourceVersion wins)
            server_rv = {
                name: api.store.pods[("default", name)]["metadata"][
                    "resourceVersion"
                ]
                for name in live
            }
            inf_rv = {
                p["metadata"]["name"]: p["metadata"]["resourceVersion"]
                for p in inf.pods()
            }
            assert wait_for(lambda: {
                p["metadata"]["name"]: p["metadata"]["resourceVersion"]
                for p in inf.pods()
            } == server_rv), f"rv mismatch: {inf_rv} != {server_rv}"
        finally:
            inf.stop()


class TestWatchCycling:
    def test_informer_survives_rapid_stream_cycling(self, api):
        """Real apiservers close watches periodically (timeoutSeconds);
        the informer must stay convergent when the stream is cycled many
        times while pods churn."""
        inf = make_informer(api, reconnect_backoff=0.01).start()
        assert inf.wait_synced(5)
        try:
            live = set()
            for i in range(10):
                name = f"cyc{i}"
                api.store.add_pod(make_pod(name, node=NODE, mem=4))
                live.add(name)
                if i % 2 == 0:
                    api.store.watch_close_all()   # server-side watch close
                if i % 3 == 2:
                    victim = sorted(live)[0]
                    api.store.delete_pod("default", victim)
                    live.discard(victim)
            assert wait_for(
                lambda: {p["metadata"]["name"] for p in inf.pods()} == live,
                timeout=15,
            )
            assert inf.reconnects >= 1
        finally:
            inf.stop()
