"""kubeshare-aggregator — cluster-wide pod GPU-demand exporter.

One `gpu_requirement` sample per running shared-GPU pod, demand in the
labels (reference pkg/aggregator/aggregator.go:22-39, pod.go:50-154:
uuid/port are read back from the env the scheduler injected).

The pod source is injected: a k8s client shim in production, the fake
cluster in tests. Each source yields PodDemand records.
"""
from __future__ import annotations

import time
from dataclasses import dataclass

from prometheus_client.core import GaugeMetricFamily

from ..utils import constants as C


@dataclass
class PodDemand:
    namespace: str
    name: str
    pod_id: str
    node: str
    uuid: str
    limit: float
    request: float
    memory: int
    port: int
    group_name: str = ""
    min_available: int = 0
    cell_id: str = ""
    lease_ms: int = 0   # sharedgpu/lease_ms latency class


def demand_from_pod(pod) -> PodDemand | None:
    """Extract a PodDemand from a scheduled FakePod / pod-like object
    (labels+annotations+env), mirroring aggregator/pod.go:74-154."""
    labels = getattr(pod, "labels", {}) or {}
    ann = getattr(pod, "annotations", {}) or {}
    env = getattr(pod, "env", {}) or {}
    if C.POD_GPU_UUID not in ann:
        return None
    try:
        # scheduler-written annotations are trusted, but a hand-crafted
        # pod can carry junk in any numeric field — one bad pod must
        # not fail the whole cluster scrape
        return PodDemand(
            namespace=pod.namespace, name=pod.name,
            pod_id=getattr(pod, "uid", ""),
            node=getattr(pod, "node", "") or getattr(pod, "node_name",
                                                     ""),
            uuid=ann.get(C.POD_GPU_UUID, ""),
            limit=float(labels.get(C.POD_GPU_LIMIT, "0") or 0),
            request=float(labels.get(C.POD_GPU_REQUEST, "0") or 0),
            memory=int(ann.get(C.POD_GPU_MEMORY, "0") or 0),
            port=int(ann.get(C.POD_MANAGER_PORT, "0") or
                     env.get(C.ENV_POD_MANAGER_PORT, "0") or 0),
            group_name=labels.get(C.POD_GROUP_NAME, ""),
            min_available=int(labels.get(C.POD_MIN_AVAILABLE, "0") or 0),
            cell_id=ann.get(C.POD_CELL_ID, ""),
            lease_ms=int(labels.get(C.POD_LEASE_MS, "0") or 0),
        )
    except ValueError:
        return None


class GPURequirementCollector:
    def __init__(self, pod_source):
        """pod_source: callable returning an iterable of PodDemand."""
        self.pod_source = pod_source

    def collect(self):
        fam = GaugeMetricFamily(
            C.METRIC_GPU_REQUIREMENT,
            "GPU demand of running shared-GPU pods",
            labels=["namespace", "pod", "pod_id", "node", "group_name",
                    "min_available", "limit", "request", "memory",
                    "cell_id", "uuid", "port", "lease_ms"])
        now = time.time()
        for d in self.pod_source():
            fam.add_metric(
                [d.namespace, d.name, d.pod_id, d.node, d.group_name,
                 str(d.min_available), str(d.limit), str(d.request),
                 str(d.memory), d.cell_id, d.uuid, str(d.port),
                 str(d.lease_ms)], now)
        yield fam


def serve(pod_source, port: int = C.AGGREGATOR_PORT):
    from prometheus_client import CollectorRegistry, start_http_server

    registry = CollectorRegistry()
    registry.register(GPURequirementCollector(pod_source))
    start_http_server(port, registry=registry)
    return registry
