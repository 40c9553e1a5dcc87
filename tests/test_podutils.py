"""Annotation protocol (scheduler-extender handshake) tests."""

from gpushare_amd import consts
from gpushare_amd.cluster import podutils

from helpers import make_pod, plain_pod


def test_gpu_memory_sums_containers():
    pod = make_pod("p", mem=6, containers=[2, 4])
    assert podutils.gpu_memory_of_pod(pod) == 6
    assert podutils.gpu_memory_of_container(pod["spec"]["containers"][1]) == 4


def test_gpu_memory_zero_for_plain_pod():
    assert podutils.gpu_memory_of_pod(plain_pod("x")) == 0


def test_gpu_id_annotation():
    assert podutils.gpu_id_from_annotation(make_pod("p", 2, gpu_idx=5)) == 5
    pod = make_pod("p", 2)
    pod["metadata"]["annotations"][consts.ENV_RESOURCE_INDEX] = "bogus"
    assert podutils.gpu_id_from_annotation(pod) == -1
    assert podutils.gpu_id_from_annotation(plain_pod("x")) == -1


def test_assume_time():
    pod = make_pod("p", 2, assume_time_ns=12345)
    assert podutils.assume_time_from_annotation(pod) == 12345
    assert podutils.assume_time_from_annotation(plain_pod("x")) == 0


def test_is_assumed_pod():
    assert podutils.is_assumed_pod(make_pod("p", 2))
    # already assigned
    assert not podutils.is_assumed_pod(make_pod("p", 2, assigned="true"))
    # no gpu-mem limit
    assert not podutils.is_assumed_pod(plain_pod("x"))
    # missing assume time
    pod = make_pod("p", 2)
    del pod["metadata"]["annotations"][consts.ENV_RESOURCE_ASSUME_TIME]
    assert not podutils.is_assumed_pod(pod)
    # missing assigned flag entirely (reference: warning, not assumed)
    pod = make_pod("p", 2)
    del pod["metadata"]["annotations"][consts.ENV_ASSIGNED_FLAG]
    assert not podutils.is_assumed_pod(pod)


def test_assigned_patch_shape():
    patch = podutils.assigned_patch(now_ns=777)
    anns = patch["metadata"]["annotations"]
    assert anns[consts.ENV_ASSIGNED_FLAG] == "true"
    assert anns[consts.ENV_RESOURCE_ASSUME_TIME] == "777"


def test_pod_is_not_running():
    assert podutils.pod_is_not_running(plain_pod("x", phase="Failed"))
    assert podutils.pod_is_not_running(plain_pod("x", phase="Succeeded"))
    assert not podutils.pod_is_not_running(plain_pod("x", phase="Running"))
    # deletionTimestamp
    pod = plain_pod("x")
    pod["metadata"]["deletionTimestamp"] = "2026-01-01T00:00:00Z"
    assert podutils.pod_is_not_running(pod)
    # Pending with only PodScheduled=True condition
    pod = plain_pod("x", phase="Pending")
    pod["status"]["conditions"] = [{"type": "PodScheduled", "status": "True"}]
    assert podutils.pod_is_not_running(pod)
    # Pending with more conditions -> running path
    pod["status"]["conditions"].append({"type": "Initialized", "status": "True"})
    assert not podutils.pod_is_not_running(pod)


def test_allocation_map_annotation():
    pod = make_pod(
        "p",
        4,
        extra_annotations={
            consts.ANN_GPUSHARE_ALLOCATION: '{"c0": {"2": 4}}'
        },
    )
    assert podutils.allocation_map_from_annotation(pod) == {"c0": {"2": 4}}
    pod["metadata"]["annotations"][consts.ANN_GPUSHARE_ALLOCATION] = "not-json"
    assert podutils.allocation_map_from_annotation(pod) is None


class TestAdversarialAnnotations:
    """Annotations are tenant-influenced cluster data: every parser must
    degrade safely on garbage (hypothesis fuzz + known nasty cases)."""

    NASTY = [
        None, "", " ", "NaN", "1e309", "-0", "0x10", "۱۲۳",  # non-ASCII digits
        "9" * 400, "[]", "{}", '{"a":', "\x00", "true", "null", "-1",
    ]

    def test_gpu_id_never_raises(self):
        for val in self.NASTY:
            pod = {"metadata": {"annotations": {}}}
            if val is not None:
                pod["metadata"]["annotations"][
                    consts.ENV_RESOURCE_INDEX
                ] = val
            idx = podutils.gpu_id_from_annotation(pod)
            assert isinstance(idx, int)

    def test_assume_time_never_raises(self):
        for val in self.NASTY:
            pod = {"metadata": {"annotations": {}}}
            if val is not None:
                pod["metadata"]["annotations"][
                    consts.ENV_RESOURCE_ASSUME_TIME
                ] = val
            t = podutils.assume_time_from_annotation(pod)
            assert isinstance(t, int)

    def test_allocation_map_never_raises(self):
        for val in self.NASTY + [
            '{"c": "notdict"}',
            '{"c": {"x": 1}}',           # non-int GPU key
            '{"c": {"0": "NaN"}}',       # non-int units
            '{"c": {"0": [1]}}',
            '[1,2,3]',
            '{"c": {"0": 1e99}}',
        ]:
            pod = {"metadata": {"annotations": {}}}
            if val is not None:
                pod["metadata"]["annotations"][
                    consts.ANN_GPUSHARE_ALLOCATION
                ] = val
            split = podutils.gpu_split_from_pod(pod)
            assert split is None or (
                isinstance(split, dict)
                and all(
                    isinstance(k, int) and isinstance(v, int)
                    for k, v in split.items()
                )
            )

    def test_is_assumed_never_raises_on_garbage(self):
        for val in self.NASTY:
            pod = {
                "metadata": {
                    "annotations": {
                        consts.ENV_RESOURCE_ASSUME_TIME: val or "",
                        consts.ENV_ASSIGNED_FLAG: val or "",
                    }
                },
                "spec": {
                    "containers": [
                        {"resources": {"limits": {consts.RESOURCE_NAME: "4"}}}
                    ]
                },
                "status": {"phase": "Pending"},
            }
            assert podutils.is_assumed_pod(pod) in (True, False)

    def test_hypothesis_fuzz_annotations(self):
        try:
            from hypothesis import given, settings, strategies as st
        except ImportError:
            import pytest

            pytest.skip("hypothesis unavailable")

        @settings(max_examples=300, deadline=None)
        @given(st.text(max_size=80))
        def run(val):
            pod = {
                "metadata": {
                    "annotations": {
                        consts.ENV_RESOURCE_INDEX: val,
                        consts.ENV_RESOURCE_ASSUME_TIME: val,
                        consts.ANN_GPUSHARE_ALLOCATION: val,
                    }
                }
            }
            podutils.gpu_id_from_annotation(pod)
            podutils.assume_time_from_annotation(pod)
            podutils.gpu_split_from_pod(pod)

        run()


def test_split_rejects_absurd_indices_and_units():
    from helpers import make_pod
    import json as _json

    for bad in (
        {"c": {"0": 10**19}},          # poisoned units
        {"c": {"0": 0}},               # zero units
        {"c": {"0": -4}},              # negative
        {"c": {"99999": 4}},           # absurd GPU index
        {"c": {"-1": 4}},
    ):
        pod = make_pod(
            "p", 4,
            extra_annotations={
                consts.ANN_GPUSHARE_ALLOCATION: _json.dumps(bad)
            },
        )
        assert podutils.gpu_split_from_pod(pod) is None, bad


def test_malformed_pod_shapes_degrade_cleanly():
    """/pods is untyped JSON; structurally-broken pods must parse as
    'not a gpushare pod', never crash the annotation accessors."""
    shapes = [
        {},
        {"metadata": None},
        {"metadata": []},
        {"metadata": "x"},
        {"metadata": {"annotations": []}},
        {"metadata": {"annotations": "notdict"}},
        {"metadata": {"annotations": None}},
    ]
    for pod in shapes:
        assert podutils.annotations(pod) == {}
        assert podutils.gpu_id_from_annotation(pod) == -1
        assert podutils.assume_time_from_annotation(pod) == 0
        assert podutils.gpu_split_from_pod(pod) is None
