"""Node daemon (L2): supervises the per-GPU native isolation stack —
one gpu-schd per GPU, one pod-mgr per sharing pod (reference
gemini-scheduler launcher glue, docker/kubeshare-gemini-scheduler/),
publishes the node's GPU inventory annotation and the per-pod
quota-enforcement metrics."""
from .launcher import NodeDaemon, PodManagerSupervisor  # noqa: F401
