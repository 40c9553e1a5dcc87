"""Binpack policy + extender core + webhook server over real HTTP."""

import pytest

from gpushare_amd import consts
from gpushare_amd.cluster.kubeclient import FakeKubeClient
from gpushare_amd.extender import BinpackState, GPUShareExtender
from gpushare_amd.extender.server import ExtenderClient, ExtenderServer

from helpers import make_pod, plain_pod


def test_binpack_best_fit():
    st = BinpackState()
    st.set_node("n", [288] * 4)
    assert st.assume("n", 72) == 0
    # best-fit: GPU 0 now has 216 free -> still the tightest fit
    assert st.assume("n", 72) == 0
    assert st.assume("n", 200) == 1   # doesn't fit 0 (216 free? yes fits)...
    st2 = BinpackState()
    st2.set_node("n", [8, 8])
    st2.assume("n", 6)                 # gpu0 free=2
    assert st2.assume("n", 4) == 1     # must go to gpu1
    assert st2.assume("n", 2) == 0     # best-fit: gpu0 free=2 < gpu1 free=4
    assert st2.assume("n", 8) is None  # nothing fits


def test_binpack_demo_colocation():
    """The reference's canonical demo: 3×2 GiB pods share ONE GPU."""
    st = BinpackState()
    st.set_node("n", [16, 16])
    assert [st.assume("n", 2) for _ in range(3)] == [0, 0, 0]


def test_binpack_release_and_packing():
    st = BinpackState()
    st.set_node("n", [288] * 2)
    st.assume("n", 100)
    st.release("n", 0, 100)
    report = st.packing()
    assert report["allocated_units"] == 0
    st.assume("n", 288)
    st.assume("n", 144)
    assert st.packing()["packing_pct"] == pytest.approx(75.0)


def test_extender_assume_writes_annotation_triple():
    kube = FakeKubeClient("node-a")
    ext = GPUShareExtender(kube)
    ext.register_node("node-a", [288] * 8)
    pod = kube.add_pod(
        {
            "metadata": {"name": "p1", "namespace": "default"},
            "spec": {
                "nodeName": "node-a",
                "containers": [
                    {"resources": {"limits": {consts.RESOURCE_NAME: "72"}}}
                ],
            },
            "status": {"phase": "Pending"},
        }
    )
    idx = ext.assume(pod, "node-a")
    assert idx == 0
    anns = kube.get_pod("default", "p1")["metadata"]["annotations"]
    assert anns[consts.ENV_RESOURCE_INDEX] == "0"
    assert anns[consts.ENV_ASSIGNED_FLAG] == "false"
    assert int(anns[consts.ENV_RESOURCE_ASSUME_TIME]) > 0


def test_extender_resync_from_annotations():
    kube = FakeKubeClient("node-a")
    kube.add_pod(make_pod("existing", 100, gpu_idx=2))
    ext = GPUShareExtender(kube)
    ext.register_node("node-a", [288] * 4)
    ext.resync()
    assert ext.state.nodes["node-a"].allocated == [0, 0, 100, 0]
    # terminal pods don't count
    kube.add_pod(make_pod("dead", 50, gpu_idx=1, phase="Failed"))
    ext.resync()
    assert ext.state.nodes["node-a"].allocated[1] == 0


def test_extender_filter():
    kube = FakeKubeClient("node-a")
    ext = GPUShareExtender(kube)
    ext.register_node("node-a", [8])
    ext.register_node("node-b", [288])
    pod = make_pod("p", 72)
    assert ext.filter(pod, ["node-a", "node-b"]) == ["node-b"]
    # non-gpu pods pass through
    assert ext.filter(plain_pod("x"), ["node-a", "node-b"]) == [
        "node-a",
        "node-b",
    ]


@pytest.fixture
def webhook():
    kube = FakeKubeClient("node-a")
    ext = GPUShareExtender(kube)
    ext.register_node("node-a", [288] * 2)
    srv = ExtenderServer(ext).start()
    client = ExtenderClient(srv.url)
    yield kube, ext, client
    client.close()
    srv.stop()


def _raw_gpu_pod(name, mem):
    """Unannotated gpushare pod — what exists *before* the extender binds."""
    return {
        "metadata": {"name": name, "namespace": "default"},
        "spec": {
            "nodeName": "node-a",
            "containers": [
                {"resources": {"limits": {consts.RESOURCE_NAME: str(mem)}}}
            ],
        },
        "status": {"phase": "Pending"},
    }


def test_webhook_filter_bind_release(webhook):
    kube, ext, client = webhook
    pod = kube.add_pod(_raw_gpu_pod("w1", 200))
    # filter
    assert client.filter(pod, ["node-a"]) == ["node-a"]
    # bind
    assert client.bind("default", "w1", "node-a") == ""
    anns = kube.get_pod("default", "w1")["metadata"]["annotations"]
    assert anns[consts.ENV_ASSIGNED_FLAG] == "false"
    assert ext.packing()["allocated_units"] == 200
    # second 200-unit pod fits GPU 1
    kube.add_pod(_raw_gpu_pod("w2", 200))
    assert client.bind("default", "w2", "node-a") == ""
    # third cannot fit
    kube.add_pod(_raw_gpu_pod("w3", 200))
    err = client.bind("default", "w3", "node-a")
    assert "no GPU" in err
    # release w1 -> fits again
    client.release(kube.get_pod("default", "w1"), "node-a")
    assert client.bind("default", "w3", "node-a") == ""
    assert client.packing()["allocated_units"] == 400


def test_rest_client_against_fake_apiserver():
    """RestKubeClient (real httpx) against the HTTP fake apiserver."""
    from gpushare_amd.cluster.fakeapiserver import FakeApiServer
    from gpushare_amd.cluster.kubeclient import RestKubeClient

    api = FakeApiServer().start()
    try:
        client = RestKubeClient(base_url=api.url, token="test-token")
        node = client.get_node("node-a")
        assert node["metadata"]["name"] == "node-a"
        client.patch_node_status(
            "node-a", {"status": {"capacity": {consts.RESOURCE_COUNT: "8"}}}
        )
        assert (
            client.get_node("node-a")["status"]["capacity"][consts.RESOURCE_COUNT]
            == "8"
        )
        api.store.add_pod(make_pod("rp", 4, gpu_idx=0))
        pods = client.list_pods(
            field_selector="spec.nodeName=node-a,status.phase=Pending"
        )
        assert len(pods["items"]) == 1
        client.patch_pod("default", "rp", {"metadata": {"annotations": {"x": "y"}}})
        assert client.get_pod("default", "rp")["metadata"]["annotations"]["x"] == "y"
        client.close()
    finally:
        api.stop()


def test_pod_manager_against_http_apiserver():
    """Full PodManager stack over the HTTP fake apiserver (no kubelet)."""
    from gpushare_amd.cluster.fakeapiserver import FakeApiServer
    from gpushare_amd.cluster.kubeclient import RestKubeClient
    from gpushare_amd.cluster.podmanager import PodManager

    api = FakeApiServer().start()
    try:
        api.store.add_pod(make_pod("hp", 8, gpu_idx=1))
        client = RestKubeClient(base_url=api.url)
        pm = PodManager(
            client,
            "node-a",
            kubelet_client=None,
            query_kubelet=False,
            cache_ttl=0.0,
            apiserver_retries=0,
        )
        cands = pm.get_candidate_pods()
        assert [p["metadata"]["name"] for p in cands] == ["hp"]
        assert pm.mark_assigned(cands[0])
        assert (
            api.store.get_pod("default", "hp")["metadata"]["annotations"][
                consts.ENV_ASSIGNED_FLAG
            ]
            == "true"
        )
        client.close()
    finally:
        api.stop()


class TestPrioritize:
    def test_binpack_scoring_prefers_fuller_node(self):
        """prioritize: the node whose placement leaves less slack scores
        higher (co-location over spreading); infeasible nodes score 0."""
        from gpushare_amd.cluster.kubeclient import FakeKubeClient
        from gpushare_amd.extender.core import GPUShareExtender

        kube = FakeKubeClient(node_name="a")
        ext = GPUShareExtender(kube, resync_interval=3600)
        ext.register_node("a", [16, 16])
        ext.register_node("b", [16, 16])
        ext.register_node("tiny", [4])
        # node a partially used -> tighter fit for an 8 GiB pod
        ext.state.assume("a", 6)

        pod = make_pod("p", mem=8, node="")
        scores = {
            e["Host"]: e["Score"]
            for e in ext.prioritize(pod, ["a", "b", "tiny"])
        }
        assert scores["a"] > scores["b"]
        assert scores["tiny"] == 0       # cannot fit 8 GiB
        assert 0 <= max(scores.values()) <= 10

    def test_prioritize_over_http(self):
        from gpushare_amd.cluster.kubeclient import FakeKubeClient
        from gpushare_amd.extender.core import GPUShareExtender
        from gpushare_amd.extender.server import ExtenderClient, ExtenderServer

        kube = FakeKubeClient(node_name="a")
        ext = GPUShareExtender(kube, resync_interval=3600)
        ext.register_node("a", [16])
        server = ExtenderServer(ext).start()
        try:
            client = ExtenderClient(server.url)
            out = client.prioritize(make_pod("p", mem=8, node=""), ["a"])
            assert out == [{"Host": "a", "Score": 10}]
            client.close()
        finally:
            server.stop()


class TestSpreadPolicy:
    def test_spread_picks_most_free_gpu(self):
        from gpushare_amd.cluster.kubeclient import FakeKubeClient
        from gpushare_amd.extender.core import GPUShareExtender

        kube = FakeKubeClient(node_name="a")
        ext = GPUShareExtender(kube, resync_interval=3600, policy="spread")
        ext.register_node("a", [16, 16])
        p1 = make_pod("p1", mem=4, node="a")
        del p1["metadata"]["annotations"]
        kube.add_pod(p1)
        assert ext.assume(p1, "a") == 0
        p2 = make_pod("p2", mem=4, node="a")
        del p2["metadata"]["annotations"]
        kube.add_pod(p2)
        # binpack would co-locate on GPU 0; spread must pick GPU 1
        assert ext.assume(p2, "a") == 1

    def test_spread_scores_prefer_emptier_node(self):
        from gpushare_amd.cluster.kubeclient import FakeKubeClient
        from gpushare_amd.extender.core import GPUShareExtender

        kube = FakeKubeClient(node_name="a")
        ext = GPUShareExtender(kube, resync_interval=3600, policy="spread")
        ext.register_node("a", [16])
        ext.register_node("b", [16])
        ext.state.assume("a", 8)
        pod = make_pod("p", mem=4, node="")
        scores = {e["Host"]: e["Score"] for e in ext.prioritize(pod, ["a", "b"])}
        assert scores["b"] > scores["a"]

    def test_unknown_policy_rejected(self):
        from gpushare_amd.cluster.kubeclient import FakeKubeClient
        from gpushare_amd.extender.core import GPUShareExtender

        with pytest.raises(ValueError):
            GPUShareExtender(FakeKubeClient("a"), policy="nope")
