"""Webhook mutate/validate tests (reference pkg/webhook tests)."""
import base64
import json

from vgpu_manager_amd.util import consts
from vgpu_manager_amd.webhook.admission import (
    VGPU_SCHEDULER_NAME,
    apply_json_patch,
    handle_admission_review,
    mutate_pod,
    validate_pod,
)

from tests.test_allocator import make_pod


def test_mutate_sets_scheduler_and_defaults():
    pod = make_pod(number=1)
    patches = mutate_pod(pod)
    out = apply_json_patch(pod, patches)
    assert out["spec"]["schedulerName"] == VGPU_SCHEDULER_NAME
    ann = out["metadata"]["annotations"]
    assert ann[consts.device_scheduler_policy_ann()] == \
        consts.POLICY_BINPACK
    assert ann[consts.compute_policy_ann()] == consts.COMPUTE_FIXED


def test_mutate_skips_non_vgpu_and_ignored():
    plain = {"metadata": {}, "spec": {"containers": [
        {"name": "c", "resources": {}}]}}
    assert mutate_pod(plain) == []
    pod = make_pod(number=1,
                   ann={consts.IGNORE_WEBHOOK_ANN: "true"})
    assert mutate_pod(pod) == []


def test_mutate_keeps_existing_annotations():
    pod = make_pod(number=1, ann={
        consts.device_scheduler_policy_ann(): consts.POLICY_SPREAD})
    out = apply_json_patch(pod, mutate_pod(pod))
    assert out["metadata"]["annotations"][
        consts.device_scheduler_policy_ann()] == consts.POLICY_SPREAD


def test_mutate_dra_conversion():
    pod = make_pod(number=2, cores=100, memory=4096)
    out = apply_json_patch(pod, mutate_pod(pod, dra_mode=True))
    limits = out["spec"]["containers"][0]["resources"]["limits"]
    assert consts.vgpu_number_resource() not in limits
    assert out["spec"]["resourceClaims"][0]["name"] == "vgpu-claim"
    originals = json.loads(out["metadata"]["annotations"][
        consts.DRA_ORIGINAL_RESOURCES_ANN])
    assert originals["main"][consts.vgpu_number_resource()] == "2"


def test_validate_bounds():
    ok, _ = validate_pod(make_pod(number=1, cores=100))
    assert ok
    ok, msg = validate_pod(make_pod(number=0))
    assert not ok and "out of range" in msg
    ok, msg = validate_pod(make_pod(number=17))
    assert not ok
    ok, msg = validate_pod(make_pod(number=1, cores=150))
    assert not ok and "vgpu-cores" in msg


def test_validate_cores_without_number():
    pod = {"metadata": {}, "spec": {"containers": [{
        "name": "c", "resources": {"limits": {
            consts.vgpu_core_resource(): 50}}}]}}
    ok, msg = validate_pod(pod)
    assert not ok and "require vgpu-number" in msg


def test_validate_annotation_values():
    ok, msg = validate_pod(make_pod(
        number=1, ann={consts.topology_mode_ann(): "bogus"}))
    assert not ok and "device-topology-mode" in msg
    ok, _ = validate_pod(make_pod(
        number=1, ann={consts.topology_mode_ann(): consts.TOPO_LINK}))
    assert ok


def test_admission_review_roundtrip():
    pod = make_pod(number=1)
    review = {"request": {"uid": "u1", "object": pod}}
    out = handle_admission_review(review, mutating=True)
    assert out["response"]["allowed"] is True
    patches = json.loads(base64.b64decode(out["response"]["patch"]))
    assert any(p["path"] == "/spec/schedulerName" for p in patches)

    bad = make_pod(number=99)
    out = handle_admission_review({"request": {"uid": "u2",
                                               "object": bad}},
                                  mutating=False)
    assert out["response"]["allowed"] is False
    assert out["response"]["status"]["code"] == 400
