"""kubeshare-amd — MI355X-native fractional GPU sharing for Kubernetes.

The capabilities of NTHU-LSALAB/KubeShare 2.0 rebuilt gfx950-first:
HIP LD_PRELOAD isolation (native/), xGMI-topology-aware scheduling
(scheduler/), amdsmi telemetry (collector/, aggregator/), per-node quota
daemons (configdaemon/, noded/), and hand-written CDNA4 kernels (ops/).
See README.md and PARITY.md.
"""
__version__ = "0.2.0"
