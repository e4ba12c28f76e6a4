"""SharePod CRD support (crd/sharepod.yaml).

The reference ships the KubeShare 1.x SharePod CRD (crd/v1.yaml, group
sharedgpu.goc) alongside the 2.0 label API. Here a SharePod is a thin
wrapper: spec.{gpuRequest,gpuLimit,...} + a pod template; the
controller materializes it as a Pod carrying the sharedgpu/* labels and
spec.schedulerName=kubeshare-scheduler, then mirrors the pod's phase
into status. The conversion is a pure function (unit-tested on CPU);
the watch loop needs the kubernetes client.
"""
from __future__ import annotations

import copy

from .utils import constants as C

_FIELD_TO_LABEL = {
    "gpuRequest": C.POD_GPU_REQUEST,
    "gpuLimit": C.POD_GPU_LIMIT,
    "gpuMem": C.POD_GPU_MEMORY,
    "gpuModel": C.POD_GPU_MODEL,
    "priority": C.POD_PRIORITY,
    "leaseMs": C.POD_LEASE_MS,
    "groupName": C.POD_GROUP_NAME,
    "groupHeadcount": C.POD_GROUP_HEADCOUNT,
    "groupThreshold": C.POD_GROUP_THRESHOLD,
}


def sharepod_to_pod(obj: dict) -> dict:
    """SharePod (as a dict, e.g. from the dynamic client) -> Pod dict."""
    meta = obj.get("metadata", {})
    spec = obj.get("spec", {})
    template = copy.deepcopy(spec.get("template", {}))
    pod = {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": template.get("metadata", {}),
        "spec": template.get("spec", {}),
    }
    pod["metadata"].setdefault("name", meta.get("name"))
    pod["metadata"].setdefault("namespace", meta.get("namespace", "default"))
    labels = pod["metadata"].setdefault("labels", {})
    for field, label in _FIELD_TO_LABEL.items():
        if field in spec and spec[field] not in (None, ""):
            labels[label] = str(spec[field])
    pod["metadata"].setdefault("ownerReferences", [{
        "apiVersion": obj.get("apiVersion", "sharedgpu.kubeshare.amd/v1"),
        "kind": obj.get("kind", "SharePod"),
        "name": meta.get("name"),
        "uid": meta.get("uid", ""),
        "controller": True,
    }])
    pod["spec"]["schedulerName"] = C.SCHEDULER_NAME
    return pod


class SharePodController:
    """Materialize SharePods as Pods and mirror status back."""

    GROUP, VERSION, PLURAL = "sharedgpu.kubeshare.amd", "v1", "sharepods"

    def __init__(self, v1=None, crd=None):
        if v1 is not None and crd is not None:
            self.v1, self.crd = v1, crd
            return
        try:
            from kubernetes import client, config
            try:
                config.load_incluster_config()
            except Exception:  # noqa: BLE001
                config.load_kube_config()
            self.v1 = client.CoreV1Api()
            self.crd = client.CustomObjectsApi()
        except ImportError:
            from .scheduler.kubeclient import RestCoreV1, RestCustomObjects
            self.v1 = RestCoreV1()
            self.crd = RestCustomObjects(self.v1)

    def reconcile_once(self):
        objs = self.crd.list_cluster_custom_object(
            self.GROUP, self.VERSION, self.PLURAL).get("items", [])
        for obj in objs:
            meta = obj["metadata"]
            ns, name = meta.get("namespace", "default"), meta["name"]
            try:
                pod = self.v1.read_namespaced_pod(name, ns)
            except Exception:  # noqa: BLE001
                self.v1.create_namespaced_pod(ns, sharepod_to_pod(obj))
                pod = None
            phase = "Creating"
            node = ""
            if pod is not None:
                phase = (pod.status.phase if pod.status else None) \
                    or "Pending"
                node = pod.spec.node_name or ""
            status = {"phase": phase, "node": node, "podName": name}
            cur = obj.get("status") or {}
            if all(cur.get(k) == v for k, v in status.items()):
                continue  # steady state: no no-op write every cycle
            self.crd.patch_namespaced_custom_object_status(
                self.GROUP, self.VERSION, ns, self.PLURAL, name,
                {"status": status})

    def run(self, interval: float = 2.0):
        import time
        while True:
            try:
                self.reconcile_once()
            except Exception as e:  # noqa: BLE001
                print(f"[sharepod] reconcile error: {e}", flush=True)
            time.sleep(interval)


if __name__ == "__main__":
    SharePodController().run()
