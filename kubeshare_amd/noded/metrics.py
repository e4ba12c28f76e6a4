"""Quota-enforcement observability: exports each local gpu-schd's STATS
as Prometheus metrics (per-pod window usage, busy share, grants), so
operators can see the actual enforced split next to the demand
(gpu_requirement) and inventory (gpu_capacity) series. The reference
only had Gemini's file logs here (SURVEY.md §5 tracing: none)."""
from __future__ import annotations

from prometheus_client.core import CounterMetricFamily, GaugeMetricFamily

from ..isolation.client import query_stats


class GpuSchdCollector:
    """One collector over all of a node's gpu-schd daemons.

    endpoints: {gpu_uuid: (host, port)}.
    """

    def __init__(self, endpoints: dict, node_name: str = ""):
        self.endpoints = endpoints
        self.node_name = node_name

    def collect(self):
        usage = GaugeMetricFamily(
            "gpu_pod_window_usage_ms",
            "pod GPU time inside the sliding window",
            labels=["node", "uuid", "pod"])
        share = GaugeMetricFamily(
            "gpu_pod_busy_share",
            "pod share of the GPU's busy time in the window",
            labels=["node", "uuid", "pod"])
        total = CounterMetricFamily(
            "gpu_pod_total_used_ms",
            "cumulative pod GPU time", labels=["node", "uuid", "pod"])
        grants = CounterMetricFamily(
            "gpu_pod_token_grants",
            "cumulative token grants", labels=["node", "uuid", "pod"])
        sampler = GaugeMetricFamily(
            "gpu_schd_busy_sampler",
            "1 = leases charged server-sampled GPU-busy, 0 = wall/RET",
            labels=["node", "uuid"])
        other = CounterMetricFamily(
            "gpu_schd_unattributed_busy_ms",
            "sampled GPU-busy with no token holder (exempt RCCL "
            "kernels, ungated processes)", labels=["node", "uuid"])
        waiters = GaugeMetricFamily(
            "gpu_schd_waiters", "token requests currently queued",
            labels=["node", "uuid"])
        revokes = CounterMetricFamily(
            "gpu_schd_revokes",
            "liveness revocations (holder died or hung)",
            labels=["node", "uuid"])
        for uuid, (host, port) in self.endpoints.items():
            try:
                st = query_stats(host, port, timeout=2.0)
            except (OSError, ValueError):
                # daemon down, or truncated/garbled STATS mid-restart:
                # skip this GPU, never fail the whole node scrape
                continue
            sampler.add_metric([self.node_name, uuid],
                               1.0 if st.get("sampler") else 0.0)
            other.add_metric([self.node_name, uuid],
                             st.get("other_busy_ms", 0.0))
            waiters.add_metric([self.node_name, uuid],
                               st.get("waiters", 0))
            revokes.add_metric([self.node_name, uuid],
                               st.get("revokes", 0))
            for pod, v in st.get("pods", {}).items():
                # .get(): a gpu-schd from an older build mid-rolling-
                # upgrade may lack newer per-pod fields
                lab = [self.node_name, uuid, pod]
                usage.add_metric(lab, v.get("usage_ms", 0.0))
                share.add_metric(lab, v.get("busy_share", 0.0))
                total.add_metric(lab, v.get("total_used_ms", 0.0))
                grants.add_metric(lab, v.get("grants", 0))
        yield usage
        yield share
        yield total
        yield grants
        yield sampler
        yield other
        yield waiters
        yield revokes


def serve(endpoints: dict, node_name: str, port: int):
    from prometheus_client import CollectorRegistry, start_http_server

    registry = CollectorRegistry()
    registry.register(GpuSchdCollector(endpoints, node_name))
    start_http_server(port, registry=registry)
    return registry
