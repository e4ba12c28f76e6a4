"""Run the aggregator against the Kubernetes API (pods scheduled by
kubeshare-scheduler, reference aggregator/pod.go:50-72)."""
import signal

from . import PodDemand, serve
from ..utils import constants as C


def kube_pod_source(api=None):
    if api is None:
        from ..scheduler.kube import make_client
        v1 = make_client()
    else:
        v1 = api

    def source():
        out = []
        pods = v1.list_pod_for_all_namespaces(
            field_selector="status.phase=Running").items
        for p in pods:
            if p.spec.scheduler_name != C.SCHEDULER_NAME:
                continue
            ann = p.metadata.annotations or {}
            labels = p.metadata.labels or {}
            if C.POD_GPU_UUID not in ann:
                continue
            try:
                out.append(PodDemand(
                    namespace=p.metadata.namespace, name=p.metadata.name,
                    pod_id=p.metadata.uid, node=p.spec.node_name or "",
                    uuid=ann.get(C.POD_GPU_UUID, ""),
                    limit=float(labels.get(C.POD_GPU_LIMIT, 0) or 0),
                    request=float(labels.get(C.POD_GPU_REQUEST, 0) or 0),
                    memory=int(ann.get(C.POD_GPU_MEMORY, 0) or 0),
                    port=int(ann.get(C.POD_MANAGER_PORT, 0) or 0),
                    group_name=labels.get(C.POD_GROUP_NAME, ""),
                    min_available=int(labels.get(C.POD_MIN_AVAILABLE, 0)
                                      or 0),
                    lease_ms=int(labels.get(C.POD_LEASE_MS, 0) or 0),
                    cell_id=ann.get(C.POD_CELL_ID, "")))
            except (ValueError, TypeError):
                continue
        return out

    return source


if __name__ == "__main__":
    serve(kube_pod_source())
    signal.pause()
