"""kubeshare-config node daemon (L2): per-GPU-UUID quota and
pod-manager-port files — the on-disk contract between the control
plane and the native isolation daemons (reference pkg/config,
query.go:70-105). Demand comes straight from the API server (daemon.py
direct feed) with a Prometheus parity source kept for reference
compatibility."""
from .daemon import ConfigDaemon  # noqa: F401
from .files import PodPort, PodQuota, read_gpu_config  # noqa: F401
from .files import write_gpu_config, write_port_config  # noqa: F401
