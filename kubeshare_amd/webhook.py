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


def shared_pod_env(namespace: str, name: str, port: str,
                   gpu_mem: str, index: str) -> list:
    """The isolation env block for one shared-GPU container (ROCm-native
    equivalent of the reference's injection, pod.go:445-457). Transport:
    UDS through the /kubeshare/sock hostPath by default; POD_MANAGER_IP
    resolves to the NODE's IP via the downward API (the pod-mgr runs in
    the hostNetwork node daemon — 127.0.0.1 inside a pod-network
    container is the container itself, not the node)."""
    return [
        {"name": C.ENV_INJECTED, "value": "1"},
        {"name": C.ENV_ROCR_VISIBLE_DEVICES, "value": index},
        {"name": C.ENV_LD_PRELOAD, "value": C.HOOK_SO_PATH},
        {"name": C.ENV_POD_MANAGER_UDS,
         "value": C.pod_manager_uds(int(port))},
        {"name": C.ENV_POD_MANAGER_IP,
         "valueFrom": {"fieldRef": {"fieldPath": "status.hostIP"}}},
        {"name": C.ENV_POD_MANAGER_PORT, "value": port},
        {"name": C.ENV_POD_NAME, "value": f"{namespace}/{name}"},
        {"name": C.ENV_GPU_MEM, "value": gpu_mem},
        {"name": C.ENV_REQUIRE_HOOK, "value": "1"},
    ]


# volumes a shared-GPU pod needs: the hook .so (read-only) and the
# pod-mgr socket dir (RW — connect() needs write access to the inode)
SHARED_MOUNTS = [
    {"name": "kubeshare-library", "mountPath": C.LIBRARY_PATH,
     "readOnly": True},
    {"name": "kubeshare-sock", "mountPath": C.SOCK_DIR},
]
SHARED_VOLUMES = [
    {"name": "kubeshare-library", "hostPath": {"path": C.LIBRARY_PATH}},
    {"name": "kubeshare-sock",
     "hostPath": {"path": C.SOCK_DIR, "type": "DirectoryOrCreate"}},
]


def build_patch(pod: dict) -> list:
    """JSONPatch for one pod dict; [] when not a shared-GPU pod or
    already injected.

    Idempotency is keyed on the KUBESHARE_INJECTED marker — NOT on the
    presence of ROCR_VISIBLE_DEVICES: a user-set device-visibility env
    must not suppress injection (that would schedule the pod onto a
    shared GPU with no hook, no pinning, no memory cap). Conflicting
    user-set ROCR/HIP_VISIBLE_DEVICES entries are replaced in place."""
    meta = pod.get("metadata", {})
    ann = meta.get("annotations") or {}
    uuid = ann.get(C.POD_GPU_UUID)
    if not uuid:
        return []
    port = ann.get(C.POD_MANAGER_PORT, "")
    gpu_mem = ann.get(C.POD_GPU_MEMORY, "0")
    index = ann.get(C.POD_GPU_INDEX, "")
    shared = bool(port)  # whole-GPU pods get no isolation layer

    if shared:
        env = shared_pod_env(meta.get("namespace", "default"),
                             meta.get("name", ""), port, gpu_mem, index)
    else:
        env = [{"name": C.ENV_INJECTED, "value": "1"},
               {"name": C.ENV_ROCR_VISIBLE_DEVICES,
                "value": index if index else uuid}]

    ours = {e["name"] for e in env} | {C.ENV_HIP_VISIBLE_DEVICES}
    patch = []
    injected_any = False
    containers = pod.get("spec", {}).get("containers", [])
    for i, container in enumerate(containers):
        existing = container.get("env") or []
        if any(e.get("name") == C.ENV_INJECTED for e in existing):
            continue  # already injected (webhook re-invocation)
        injected_any = True
        # replace conflicting user-set entries in place, append the rest
        conflict_idx = {e.get("name"): j for j, e in enumerate(existing)
                        if e.get("name") in ours}
        if not existing:
            patch.append({"op": "add", "path": f"/spec/containers/{i}/env",
                          "value": env})
        else:
            replaced = set()
            for e in env:
                j = conflict_idx.get(e["name"])
                if j is not None:
                    replaced.add(e["name"])
                    patch.append({"op": "replace",
                                  "path": f"/spec/containers/{i}/env/{j}",
                                  "value": e})
                else:
                    patch.append({"op": "add",
                                  "path": f"/spec/containers/{i}/env/-",
                                  "value": e})
            # conflicting names we do NOT inject ourselves (a user-set
            # HIP_VISIBLE_DEVICES would filter against the ROCR-pinned
            # view and hide the device): neutralize in place
            for name, j in conflict_idx.items():
                if name not in replaced:
                    patch.append({"op": "replace",
                                  "path": f"/spec/containers/{i}/env/{j}",
                                  "value": {"name": name, "value": "0"}})
        if shared:
            mounts = container.get("volumeMounts")
            have = {m.get("name") for m in mounts or []}
            want = [m for m in SHARED_MOUNTS if m["name"] not in have]
            if mounts is None:
                patch.append({"op": "add",
                              "path": f"/spec/containers/{i}/volumeMounts",
                              "value": want})
            else:
                for m in want:
                    patch.append({"op": "add",
                                  "path": f"/spec/containers/{i}/volumeMounts/-",
                                  "value": m})
    if shared and injected_any:
        vols = pod.get("spec", {}).get("volumes")
        have = {v.get("name") for v in vols or []}
        want = [v for v in SHARED_VOLUMES if v["name"] not in have]
        if vols is None:
            patch.append({"op": "add", "path": "/spec/volumes",
                          "value": want})
        else:
            for v in want:
                patch.append({"op": "add", "path": "/spec/volumes/-",
                              "value": v})
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
