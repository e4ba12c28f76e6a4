"""kubeshare-collector — per-node GPU inventory exporter.

Exports one `gpu_capacity` sample per physical GPU with inventory in the
labels and the scrape unix-time as the value (reference
pkg/collector/collector.go:42-60; NVML swapped for amdsmi). Extra
MI355X-native label: xgmi_links (the node-local link count) so the
cluster scheduler can rebuild the link graph without a node round-trip.

Run:  python -m kubeshare_amd.collector --port 9004
"""
from __future__ import annotations

import time

from prometheus_client.core import GaugeMetricFamily

from ..utils import constants as C


class GPUCapacityCollector:
    """prometheus_client custom collector over an inventory provider
    (amdsmi on a real node; any object with local_gpus() in tests)."""

    def __init__(self, node_name: str, provider):
        self.node_name = node_name
        self.provider = provider

    def collect(self):
        fam = GaugeMetricFamily(
            C.METRIC_GPU_CAPACITY,
            "physical GPU inventory of a SharedGPU node",
            labels=["node", "uuid", "model", "memory", "index",
                    "xgmi_links"])
        now = time.time()
        for gpu in self.provider.local_gpus():
            # model names: spaces -> dashes (reference collector/gpu.go:60)
            model = gpu.model.replace(" ", "-")
            fam.add_metric(
                [self.node_name, gpu.uuid, model, str(gpu.memory),
                 str(gpu.index), str(len(gpu.xgmi_links))], now)
        yield fam


def serve(node_name: str, provider=None, port: int = C.COLLECTOR_PORT):
    from prometheus_client import CollectorRegistry, start_http_server

    if provider is None:
        from ..scheduler.inventory import AmdSmiInventory
        provider = AmdSmiInventory()
    registry = CollectorRegistry()
    registry.register(GPUCapacityCollector(node_name, provider))
    start_http_server(port, registry=registry)
    return registry


def main():
    import argparse
    import os
    import signal

    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=C.COLLECTOR_PORT)
    ap.add_argument("--node", default=os.environ.get("NODE_NAME", ""))
    args = ap.parse_args()
    serve(args.node or os.uname().nodename, port=args.port)
    signal.pause()


if __name__ == "__main__":
    main()
