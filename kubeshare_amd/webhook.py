"""Mutating admission webhook — the shadow-pod-free injection path.

The reference applies its env injection by deleting and recreating the
scheduled pod ("shadow pod", pkg/scheduler/scheduler.go:515-528), which
churns the pod UID and races controllers. The cleaner flow (SURVEY.md
§7 phase 1.3): the scheduler only writes annotations at Reserve; THIS
webhook intercepts pod CREATE/UPDATE and, when the kubeshare
annotations are present, injects the env block and the hostPath mount
in-place — no delete/recreate.

Serve:  uvicorn kubeshare_amd.webhook:app --port 8443 \
            --ssl-keyfile ... --ssl-certfile ...
Register with a MutatingWebhookConfiguration on pods (CREATE), scoped
to objects carrying the sharedgpu/gpu_uuid annotation.

The mutation itself (build_patch) is a pure function over the pod dict
— unit-tested on CPU.
"""
from __future__ import annotations

import base64
import json

from .utils import constants as C


def build_patch(pod: dict) -> list:
    """JSONPatch for one pod dict; [] when not a shared-GPU pod or
    already injected."""
    meta = pod.get("metadata", {})
    ann = meta.get("annotations") or {}
    uuid = ann.get(C.POD_GPU_UUID)
    if not uuid:
        return []
    port = ann.get(C.POD_MANAGER_PORT, "")
    gpu_mem = ann.get(C.POD_GPU_MEMORY, "0")
    index = ann.get(C.POD_GPU_INDEX, "")
    shared = bool(port)  # whole-GPU pods get no isolation layer

    env = [{"name": C.ENV_ROCR_VISIBLE_DEVICES,
            "value": index if index else uuid}]
    if shared:
        env += [
            {"name": C.ENV_LD_PRELOAD, "value": C.HOOK_SO_PATH},
            {"name": C.ENV_POD_MANAGER_IP, "value": "127.0.0.1"},
            {"name": C.ENV_POD_MANAGER_PORT, "value": port},
            {"name": C.ENV_POD_NAME,
             "value": f"{meta.get('namespace', 'default')}/{meta.get('name')}"},
            {"name": C.ENV_GPU_MEM, "value": gpu_mem},
            {"name": C.ENV_REQUIRE_HOOK, "value": "1"},
        ]

    patch = []
    containers = pod.get("spec", {}).get("containers", [])
    for i, container in enumerate(containers):
        existing = {e.get("name") for e in container.get("env") or []}
        if C.ENV_ROCR_VISIBLE_DEVICES in existing:
            continue  # already injected
        if container.get("env") is None:
            patch.append({"op": "add", "path": f"/spec/containers/{i}/env",
                          "value": env})
        else:
            for e in env:
                patch.append({"op": "add",
                              "path": f"/spec/containers/{i}/env/-",
                              "value": e})
        if shared:
            mount = {"name": "kubeshare-library",
                     "mountPath": C.LIBRARY_PATH, "readOnly": True}
            if container.get("volumeMounts") is None:
                patch.append({"op": "add",
                              "path": f"/spec/containers/{i}/volumeMounts",
                              "value": [mount]})
            else:
                patch.append({"op": "add",
                              "path": f"/spec/containers/{i}/volumeMounts/-",
                              "value": mount})
    if shared and patch:
        vol = {"name": "kubeshare-library",
               "hostPath": {"path": C.LIBRARY_PATH}}
        vols = pod.get("spec", {}).get("volumes")
        if vols is None:
            patch.append({"op": "add", "path": "/spec/volumes",
                          "value": [vol]})
        elif not any(v.get("name") == "kubeshare-library" for v in vols):
            patch.append({"op": "add", "path": "/spec/volumes/-",
                          "value": vol})
    return patch


def admission_response(review: dict) -> dict:
    request = review.get("request", {})
    pod = request.get("object", {})
    patch = build_patch(pod)
    resp = {"uid": request.get("uid"), "allowed": True}
    if patch:
        resp["patchType"] = "JSONPatch"
        resp["patch"] = base64.b64encode(
            json.dumps(patch).encode()).decode()
    return {"apiVersion": review.get("apiVersion",
                                     "admission.k8s.io/v1"),
            "kind": "AdmissionReview", "response": resp}


def make_app():
    from fastapi import FastAPI

    app = FastAPI(title="kubeshare-amd mutating webhook")

    @app.post("/mutate")
    async def mutate(review: dict):
        return admission_response(review)

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    return app


try:  # uvicorn entry point: kubeshare_amd.webhook:app
    app = make_app()
except ImportError:  # fastapi optional for the rest of the package
    app = None
