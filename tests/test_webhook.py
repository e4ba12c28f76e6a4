"""Mutating-webhook injection tests (the shadow-pod-free path): the
JSONPatch builder and the AdmissionReview endpoint."""
import base64
import json

import pytest

from kubeshare_amd.scheduler.harness import FakeCluster
from kubeshare_amd.utils import constants as C
from kubeshare_amd.webhook import admission_response, build_patch


def _pod_with_annotations(ann):
    return {
        "metadata": {"name": "p1", "namespace": "ns", "annotations": ann},
        "spec": {"containers": [{"name": "main", "image": "x"}]},
    }


def test_patch_for_shared_pod():
    ann = {C.POD_GPU_UUID: "GPU-1", C.POD_MANAGER_PORT: "50050",
           C.POD_GPU_MEMORY: "1000", C.POD_GPU_INDEX: "3"}
    patch = build_patch(_pod_with_annotations(ann))
    env_ops = [p for p in patch if p["path"].endswith("/env")]
    assert env_ops, patch
    env = {e["name"]: e.get("value") for e in env_ops[0]["value"]}
    assert env[C.ENV_ROCR_VISIBLE_DEVICES] == "3"
    assert env[C.ENV_LD_PRELOAD] == C.HOOK_SO_PATH
    assert env[C.ENV_POD_MANAGER_PORT] == "50050"
    assert env[C.ENV_POD_NAME] == "ns/p1"
    assert env[C.ENV_GPU_MEM] == "1000"
    assert any(p["path"] == "/spec/volumes" for p in patch)


def test_patch_for_whole_gpu_pod_no_hook():
    ann = {C.POD_GPU_UUID: "GPU-1,GPU-2", C.POD_GPU_INDEX: "1,2"}
    patch = build_patch(_pod_with_annotations(ann))
    env = {e["name"]: e.get("value") for p in patch
           if p["path"].endswith("/env") for e in p["value"]}
    assert env[C.ENV_ROCR_VISIBLE_DEVICES] == "1,2"
    assert C.ENV_LD_PRELOAD not in env
    assert not any(p["path"] == "/spec/volumes" for p in patch)


def test_patch_noop_for_regular_pod():
    assert build_patch(_pod_with_annotations({})) == []


def test_patch_idempotent():
    """Idempotency keys on the KUBESHARE_INJECTED marker, not on
    ROCR_VISIBLE_DEVICES (round-1 advisor finding: a user-set
    device-visibility env must not silently bypass the hook)."""
    ann = {C.POD_GPU_UUID: "GPU-1", C.POD_MANAGER_PORT: "50050",
           C.POD_GPU_INDEX: "0"}
    pod = _pod_with_annotations(ann)
    pod["spec"]["containers"][0]["env"] = [
        {"name": C.ENV_INJECTED, "value": "1"}]
    assert build_patch(pod) == []


def test_patch_replaces_user_set_visibility_env():
    """A container that sets ROCR_VISIBLE_DEVICES itself still gets the
    full injection — the conflicting entry is replaced in place."""
    ann = {C.POD_GPU_UUID: "GPU-1", C.POD_MANAGER_PORT: "50050",
           C.POD_GPU_MEMORY: "1000", C.POD_GPU_INDEX: "3"}
    pod = _pod_with_annotations(ann)
    pod["spec"]["containers"][0]["env"] = [
        {"name": "FOO", "value": "bar"},
        {"name": C.ENV_ROCR_VISIBLE_DEVICES, "value": "0,1,2,3"}]
    patch = build_patch(pod)
    replaces = [p for p in patch if p["op"] == "replace"]
    assert any(p["path"] == "/spec/containers/0/env/1" and
               p["value"]["value"] == "3" for p in replaces)
    env_names = [p["value"]["name"] for p in patch
                 if p["op"] == "add" and "/env/-" in p["path"]]
    assert C.ENV_LD_PRELOAD in env_names
    assert C.ENV_POD_MANAGER_UDS in env_names


def test_admission_review_roundtrip():
    ann = {C.POD_GPU_UUID: "GPU-1", C.POD_MANAGER_PORT: "50051",
           C.POD_GPU_INDEX: "0"}
    review = {"apiVersion": "admission.k8s.io/v1", "kind": "AdmissionReview",
              "request": {"uid": "rev-1",
                          "object": _pod_with_annotations(ann)}}
    out = admission_response(review)
    assert out["response"]["uid"] == "rev-1"
    assert out["response"]["allowed"] is True
    patch = json.loads(base64.b64decode(out["response"]["patch"]))
    assert patch


def test_scheduler_annotations_feed_webhook():
    """End-to-end: Reserve's annotations alone are enough for the
    webhook to reproduce the full env injection."""
    fc = FakeCluster(nodes={"node-a": {"gpus": 2}})
    pod = fc.add_pod("ns", "w1", {C.POD_GPU_REQUEST: "0.5",
                                  C.POD_GPU_LIMIT: "1.0"})
    fc.schedule_pending()
    patch = build_patch({
        "metadata": {"name": "w1", "namespace": "ns",
                     "annotations": pod.annotations},
        "spec": {"containers": [{"name": "main"}]},
    })
    env = {e["name"]: e.get("value") for p in patch
           if p["path"].endswith("/env") for e in p["value"]}
    assert env[C.ENV_POD_MANAGER_PORT] == \
        pod.annotations[C.POD_MANAGER_PORT]
    assert env[C.ENV_GPU_MEM] == pod.annotations[C.POD_GPU_MEMORY]


def test_fastapi_endpoint():
    fastapi = pytest.importorskip("fastapi")  # noqa: F841
    from fastapi.testclient import TestClient
    from kubeshare_amd.webhook import make_app
    client = TestClient(make_app())
    assert client.get("/healthz").json() == {"ok": True}
    ann = {C.POD_GPU_UUID: "GPU-1", C.POD_MANAGER_PORT: "50052",
           C.POD_GPU_INDEX: "0"}
    r = client.post("/mutate", json={
        "apiVersion": "admission.k8s.io/v1",
        "request": {"uid": "u", "object": _pod_with_annotations(ann)}})
    assert r.status_code == 200
    assert r.json()["response"]["allowed"] is True


def test_patch_application_is_idempotent():
    """Invariant: applying the webhook patch once makes a second
    invocation a no-op (re-invocation-safe MutatingWebhookConfiguration
    with reinvocationPolicy: IfNeeded)."""
    from kubeshare_amd.testing.fake_apiserver import apply_json_patch
    ann = {C.POD_GPU_UUID: "GPU-1", C.POD_MANAGER_PORT: "50055",
           C.POD_GPU_MEMORY: "123", C.POD_GPU_INDEX: "2"}
    pod = _pod_with_annotations(ann)
    pod["spec"]["containers"].append({"name": "sidecar", "env": [
        {"name": "HIP_VISIBLE_DEVICES", "value": "7"}]})
    patch1 = build_patch(pod)
    assert patch1
    mutated = apply_json_patch(pod, patch1)
    assert build_patch(mutated) == []
    # every container got the marker + ROCR pinning
    for c in mutated["spec"]["containers"]:
        names = {e["name"] for e in c["env"]}
        assert C.ENV_INJECTED in names
        assert C.ENV_ROCR_VISIBLE_DEVICES in names
    # the sidecar's conflicting HIP_VISIBLE_DEVICES was neutralized
    # (it would filter against the ROCR-pinned single-device view)
    side_env = {e["name"]: e.get("value")
                for e in mutated["spec"]["containers"][1]["env"]}
    assert side_env["HIP_VISIBLE_DEVICES"] == "0"
    assert side_env[C.ENV_ROCR_VISIBLE_DEVICES] == "2"
    # volumes present exactly once
    vols = [v["name"] for v in mutated["spec"]["volumes"]]
    assert sorted(vols) == ["kubeshare-library", "kubeshare-sock"]
