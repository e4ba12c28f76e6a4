"""kubeshare-config daemon — turns pod GPU demand into the per-UUID
files gpu-schd and the launcher consume (reference pkg/config: informer
event -> Prometheus gpu_requirement query -> convertData -> writeFile,
query.go:22-105).

MI355X-native change (SURVEY.md §7 phase 3): the primary feed is a
direct pod source (API-server informer equivalent) rather than a
Prometheus round-trip — the reference's path adds a scrape interval of
eventual-consistency lag during which a pod can start before its quota
file lands (README.md:141). A Prometheus-backed source is still
provided for drop-in parity.
"""
from __future__ import annotations

from . import files as F
from ..aggregator import PodDemand
from ..utils import constants as C


class ConfigDaemon:
    def __init__(self, node_name: str, config_dir: str, port_dir: str):
        self.node_name = node_name
        self.config_dir = config_dir
        self.port_dir = port_dir

    def update(self, demands: list[PodDemand]):
        """Rewrite the per-UUID files from this node's sharing pods.
        File format per reference query.go:70-105; fractional pods only
        (request <= 1.0; whole-GPU pods bypass isolation)."""
        by_uuid_cfg: dict[str, list] = {}
        by_uuid_port: dict[str, list] = {}
        for d in demands:
            if d.node != self.node_name or not d.uuid:
                continue
            if d.request > 1.0:
                continue
            pod = f"{d.namespace}/{d.name}"
            by_uuid_cfg.setdefault(d.uuid, []).append(
                F.PodQuota(pod, d.limit, d.request, d.memory,
                           group=d.group_name,
                           lease_ms=getattr(d, "lease_ms", 0)))
            by_uuid_port.setdefault(d.uuid, []).append(
                F.PodPort(pod, d.port))
        if not by_uuid_cfg:
            F.zero_files(self.config_dir, self.port_dir)
            return
        import os
        known = set(os.listdir(self.config_dir))
        for uuid, quotas in by_uuid_cfg.items():
            F.write_gpu_config(self.config_dir, uuid, quotas)
            F.write_port_config(self.port_dir, uuid, by_uuid_port[uuid])
            known.discard(uuid)
        # GPUs that lost their last sharing pod go back to "0"
        for uuid in known:
            if uuid.endswith(".tmp"):
                continue
            F.write_gpu_config(self.config_dir, uuid, [])
            F.write_port_config(self.port_dir, uuid, [])


class PrometheusPodSource:
    """Parity source: scrape gpu_requirement from a Prometheus server
    (reference query.go:22-65)."""

    def __init__(self, prom_url: str, node_name: str):
        self.url = prom_url.rstrip("/")
        self.node = node_name

    def __call__(self) -> list[PodDemand]:
        import requests
        r = requests.get(
            f"{self.url}/api/v1/series",
            params={"match[]":
                    f'{{__name__="{C.METRIC_GPU_REQUIREMENT}",'
                    f'node="{self.node}"}}'},
            timeout=10)
        r.raise_for_status()
        out = []
        for s in r.json().get("data", []):
            try:
                out.append(PodDemand(
                    namespace=s.get("exported_namespace",
                                    s.get("namespace", "")),
                    name=s.get("exported_pod", s.get("pod", "")),
                    pod_id=s.get("pod_id", ""), node=s.get("node", ""),
                    uuid=s.get("uuid", "").replace(",", ""),
                    limit=float(s.get("limit", 0)),
                    request=float(s.get("request", 0)),
                    memory=int(s.get("memory", 0)),
                    port=int(s.get("port", 0)),
                    group_name=s.get("group_name", ""),
                    cell_id=s.get("cell_id", ""),
                    lease_ms=int(s.get("lease_ms", 0) or 0)))
            except (ValueError, TypeError):
                continue
        return out
