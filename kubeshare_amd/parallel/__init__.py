"""DDP gang helpers — RCCL-over-xGMI workloads under the isolation
layer.

The reference schedules PyTorch Elastic DDP gangs but contains no
collective code of its own (NCCL lives in the workload
images). Here the workload side is first-class because it is the risky
interaction: RCCL collectives inside token-gated ranks (the hook
exempts librccl call sites and gpu-schd co-grants gang members so a
collective in one rank never spins on a token-starved peer —
native/hook/hiphook.cpp, native/schd/token_sched.hpp).
"""
from .ddp import ddp_worker, launch_gang  # noqa: F401
