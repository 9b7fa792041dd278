"""Annotation-contract tests: wire-format compatibility with the reference
(pkg/internal/utils.go l.172-289), including the legacy gpu*->leafCell* key
conversion (l.189-197), legacy annotation-key fallback, defaulting, and the
bind-info roundtrip that makes pods the database (SURVEY.md §5 checkpoint).
"""
import pytest
import yaml

from hivedscheduler_amd.api import constants
from hivedscheduler_amd.api.types import PodBindInfo, WebServerError
from hivedscheduler_amd.internal import pod as podmod


def make_pod(annotations=None, limits=None, name="p1", ns="ns"):
    return {
        "metadata": {"name": name, "namespace": ns, "uid": f"uid-{name}",
                     "annotations": annotations or {}},
        "spec": {"containers": [{"resources": {"limits": limits or {}}}]},
        "status": {},
    }


def test_extract_spec_modern_keys():
    ann = {constants.AnnotationKeyPodSchedulingSpec: yaml.safe_dump({
        "virtualCluster": "VC1", "priority": 10, "leafCellNumber": 4})}
    spec = podmod.extract_pod_scheduling_spec(make_pod(ann))
    assert spec.virtualCluster == "VC1"
    assert spec.leafCellNumber == 4
    # defaulting: singleton affinity group named ns/name
    assert spec.affinityGroup.name == "ns/p1"
    assert spec.affinityGroup.members[0].podNumber == 1
    assert spec.affinityGroup.members[0].leafCellNumber == 4


def test_extract_spec_legacy_gpu_keys():
    """gpuType/gpuNumber in the YAML body are rewritten to leafCell* on read
    (reference internal/utils.go:189-197)."""
    ann = {constants.AnnotationKeyPodSchedulingSpec: yaml.safe_dump({
        "virtualCluster": "VC1", "priority": 0,
        "gpuType": "MI355X", "gpuNumber": 2,
        "affinityGroup": {"name": "g", "members": [{"podNumber": 3, "gpuNumber": 2}]}})}
    spec = podmod.extract_pod_scheduling_spec(make_pod(ann))
    assert spec.leafCellType == "MI355X"
    assert spec.leafCellNumber == 2
    assert spec.affinityGroup.members[0].leafCellNumber == 2


def test_extract_spec_legacy_annotation_key():
    """The reference's hivedscheduler.microsoft.com annotation key still
    works, so existing HiveD clients can switch without edits."""
    ann = {constants.LegacyAnnotationKeyPodSchedulingSpec: yaml.safe_dump({
        "virtualCluster": "VC2", "priority": -1, "leafCellNumber": 1})}
    spec = podmod.extract_pod_scheduling_spec(make_pod(ann))
    assert spec.virtualCluster == "VC2"
    assert spec.priority == -1


@pytest.mark.parametrize("body,msg", [
    ({"priority": 0, "leafCellNumber": 1}, "VirtualCluster"),
    ({"virtualCluster": "VC1", "priority": -2, "leafCellNumber": 1}, "Priority"),
    ({"virtualCluster": "VC1", "priority": 1001, "leafCellNumber": 1}, "Priority"),
    ({"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 0}, "LeafCellNumber"),
    ({"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2,
      "affinityGroup": {"name": "g", "members": [{"podNumber": 1, "leafCellNumber": 3}]}},
     "does not contain current Pod"),
    ({"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2,
      "affinityGroup": {"name": "g", "members": [{"podNumber": 0, "leafCellNumber": 2}]}},
     "non-positive PodNumber"),
])
def test_extract_spec_validation_errors(body, msg):
    ann = {constants.AnnotationKeyPodSchedulingSpec: yaml.safe_dump(body)}
    with pytest.raises(WebServerError) as ei:
        podmod.extract_pod_scheduling_spec(make_pod(ann))
    assert msg in str(ei.value)
    assert ei.value.code == 400


def test_missing_spec_annotation_is_bad_request():
    with pytest.raises(WebServerError) as ei:
        podmod.extract_pod_scheduling_spec(make_pod({}))
    assert ei.value.code == 400


def test_bind_info_roundtrip_via_binding_pod():
    """new_binding_pod stamps node + isolation + bind-info; extract_pod_bind_info
    recovers the identical PodBindInfo (this is the crash-recovery path)."""
    info = PodBindInfo.from_dict({
        "node": "node1",
        "leafCellIsolation": [4, 5, 6, 7],
        "cellChain": "MI355X-NODE",
        "affinityGroupBindInfo": [
            {"podPlacements": [
                {"physicalNode": "node1", "physicalLeafCellIndices": [4, 5, 6, 7],
                 "preassignedCellTypes": ["MI355X-NODE"] * 4}]}],
    })
    binding = podmod.new_binding_pod(make_pod({}), info)
    assert binding["spec"]["nodeName"] == "node1"
    ann = binding["metadata"]["annotations"]
    assert ann[constants.AnnotationKeyPodLeafCellIsolation] == "4,5,6,7"
    back = podmod.extract_pod_bind_info(binding)
    assert back.to_dict() == info.to_dict()


def test_bind_info_legacy_physical_gpu_indices():
    """Legacy physicalGpuIndices inside pod-bind-info is converted on read."""
    raw = yaml.safe_dump({
        "node": "n1", "gpuIsolation": [0, 1], "cellChain": "MI355X-NODE",
        "affinityGroupBindInfo": [
            {"podPlacements": [{"physicalNode": "n1", "physicalGpuIndices": [0, 1],
                                "preassignedCellTypes": ["MI355X-NODE", "MI355X-NODE"]}]}],
    })
    pod = make_pod({constants.AnnotationKeyPodBindInfo: raw})
    info = podmod.extract_pod_bind_info(pod)
    assert info.leafCellIsolation == [0, 1]
    assert info.affinityGroupBindInfo[0].podPlacements[0].physicalLeafCellIndices == [0, 1]


def test_opt_in_resource_limits():
    """Only pods whose containers set the pod-scheduling-enable resource limit
    (modern or legacy name) are 'interested'."""
    assert not podmod.is_interested(make_pod())
    assert podmod.is_interested(
        make_pod(limits={constants.ResourceNamePodSchedulingEnable: "1"}))
    assert podmod.is_interested(
        make_pod(limits={constants.LegacyResourceNamePodSchedulingEnable: 1}))
    done = make_pod(limits={constants.ResourceNamePodSchedulingEnable: "1"})
    done["status"]["phase"] = "Succeeded"
    assert not podmod.is_interested(done)


def test_node_health_predicate():
    ready = {"spec": {}, "status": {"conditions": [{"type": "Ready", "status": "True"}]}}
    assert podmod.is_node_healthy(ready)
    assert not podmod.is_node_healthy(
        {"spec": {"unschedulable": True}, "status": ready["status"]})
    assert not podmod.is_node_healthy(
        {"spec": {}, "status": {"conditions": [{"type": "Ready", "status": "False"}]}})
