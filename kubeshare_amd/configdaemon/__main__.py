"""Node config daemon main loop: pod demand (API server) -> per-UUID
quota/port files. Direct feed, not the reference's Prometheus
round-trip (see daemon.py)."""
import argparse
import os
import time

from ..utils import constants as C
from .daemon import ConfigDaemon


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--node", default=os.environ.get("NODE_NAME", ""))
    ap.add_argument("--config-dir", default=C.GPU_CONFIG_DIR)
    ap.add_argument("--port-dir", default=C.POD_MANAGER_PORT_DIR)
    ap.add_argument("--interval", type=float, default=2.0)
    ap.add_argument("--prometheus", default="",
                    help="optional: poll gpu_requirement from this "
                         "Prometheus URL instead of the API server")
    args = ap.parse_args()
    os.makedirs(args.config_dir, exist_ok=True)
    os.makedirs(args.port_dir, exist_ok=True)
    daemon = ConfigDaemon(args.node, args.config_dir, args.port_dir)
    if args.prometheus:
        from .daemon import PrometheusPodSource
        source = PrometheusPodSource(args.prometheus, args.node)
    else:
        from ..aggregator.__main__ import kube_pod_source
        source = kube_pod_source()
    while True:
        try:
            daemon.update([d for d in source() if d.node == args.node])
        except Exception as e:  # noqa: BLE001
            print(f"[kubeshare-config] update failed: {e}", flush=True)
        time.sleep(args.interval)


if __name__ == "__main__":
    main()
