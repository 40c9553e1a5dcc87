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
