"""Pod label parsing & validation.

Reproduces the observable accept/reject behavior of the reference's
getPodLabels / getPodPrioriy (pkg/scheduler/pod.go:175-327) with a sane
value grammar (the reference regex at pod.go:20 uses an unescaped `.`;
we use a proper one — same accept set on digit strings, SURVEY.md
Appendix A).

Semantics (reference pod.go:249-305, README.md:37-45):
- fractional share: 0 < request <= limit <= 1.0
- whole GPUs:       limit > 1.0 must be an integer AND limit == request
- both 0 / labels absent  -> regular (non-shared) pod
- priority: unset -> 0 (Opportunistic); valid range -1..100; 1..100 are
  Guarantee pods (pod.go:179-199)
"""
from __future__ import annotations

import math
import re
from dataclasses import dataclass, field
from typing import Optional

from . import constants as C

# "0.5" | "2.0" (integer with .0+) | "2" (bare integer)
_VALUE_RE = re.compile(r"^(?:0+\.[0-9]+|[1-9][0-9]*\.0+|[1-9][0-9]*)$")


class LabelError(ValueError):
    """A sharedgpu/* label was present but malformed (pod must be rejected)."""


@dataclass
class PodSpec:
    """Parsed scheduling-relevant state of one pod (reference PodStatus)."""

    namespace: str
    name: str
    uid: str = ""
    node_name: str = ""
    priority: int = 0
    limit: float = 0.0
    request: float = 0.0
    memory: int = 0
    model: str = ""
    pod_group: str = ""
    min_available: int = 0
    headcount: int = 0  # sharedgpu/group_headcount (Permit timeout unit)
    lease_ms: int = 0   # sharedgpu/lease_ms latency class (0 = default)
    # filled at Reserve time
    uuids: list = field(default_factory=list)
    cell_ids: list = field(default_factory=list)
    port: int = 0

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}"

    @property
    def is_shared(self) -> bool:
        return 0.0 < self.limit <= 1.0

    @property
    def is_multi_gpu(self) -> bool:
        return self.limit > 1.0

    @property
    def is_opportunistic(self) -> bool:
        return self.priority <= 0

    def default_memory(self, full_memory: int = C.MI355X_HBM_BYTES) -> int:
        """gpu_mem default = floor(request * full GPU memory)
        (reference pod.go:419-421)."""
        if self.memory > 0:
            return self.memory
        return math.floor(self.request * full_memory)


def _parse_value(raw: str, label: str) -> float:
    if _VALUE_RE.fullmatch(raw) is None:
        raise LabelError(f"{label} set error by user: {raw!r}")
    return float(raw)


def parse_priority(labels: dict) -> int:
    raw = labels.get(C.POD_PRIORITY)
    if raw is None or raw == "":
        return 0
    try:
        p = int(raw)
    except ValueError as e:
        raise LabelError(f"{C.POD_PRIORITY} set error by user: {raw!r}") from e
    if p > 100 or p < -1:
        raise LabelError(f"{C.POD_PRIORITY} out of range [-1,100]: {p}")
    return p


def parse_gang(labels: dict) -> tuple[str, int, int]:
    """Returns (group_name, min_available, headcount).

    The scheduler vocabulary derives min_available =
    floor(headcount*threshold + 0.5) (reference pod_group.go:86-117);
    the aggregator vocabulary is a direct sharedgpu/min_available label
    (aggregator/pod.go:22,92). Both accepted; direct label wins for
    min_available. headcount is kept separately because the Permit gang
    timeout is 2 s x HEADCOUNT, not min_available (reference
    scheduler.go:44,573); 0 when the label is absent.
    """
    group = labels.get(C.POD_GROUP_NAME, "")
    headcount = 0
    head_raw = labels.get(C.POD_GROUP_HEADCOUNT)
    if head_raw is not None:
        try:
            headcount = int(head_raw)
        except ValueError as e:
            raise LabelError("group_headcount set error") from e
        if headcount < 0:
            raise LabelError("group_headcount negative")
    direct = labels.get(C.POD_MIN_AVAILABLE)
    if direct is not None:
        try:
            return group, max(0, int(direct)), headcount
        except ValueError as e:
            raise LabelError(f"{C.POD_MIN_AVAILABLE} set error: {direct!r}") from e
    thr_raw = labels.get(C.POD_GROUP_THRESHOLD)
    if head_raw is None or thr_raw is None:
        return group, 0, headcount
    try:
        threshold = float(thr_raw)
    except ValueError as e:
        raise LabelError("group_headcount/group_threshold set error") from e
    if not (0.0 <= threshold <= 1.0):
        raise LabelError("group_headcount/group_threshold out of range")
    return group, int(math.floor(headcount * threshold + 0.5)), headcount


def parse_pod(namespace: str, name: str, labels: dict, *, uid: str = "",
              node_name: str = "") -> Optional[PodSpec]:
    """Parse one pod's sharedgpu labels.

    Returns None for a regular pod (no GPU sharing labels, or limit and
    request both zero — reference pod.go:276-305). Raises LabelError when
    labels are present but invalid (the pod must be rejected, not treated
    as regular).
    """
    spec = PodSpec(namespace=namespace, name=name, uid=uid, node_name=node_name)
    spec.pod_group, spec.min_available, spec.headcount = parse_gang(labels)
    spec.priority = parse_priority(labels)

    raw_limit = labels.get(C.POD_GPU_LIMIT)
    raw_request = labels.get(C.POD_GPU_REQUEST)
    raw_memory = labels.get(C.POD_GPU_MEMORY)

    if raw_limit is None and raw_request is None and raw_memory is None:
        return None  # regular pod

    # a pod that wants GPU must set the limit label (pod.go:294-300)
    if raw_limit is None:
        raise LabelError(f"{C.POD_GPU_LIMIT} must be set for a shared-GPU pod")
    limit = _parse_value(raw_limit, C.POD_GPU_LIMIT)

    request = 0.0
    if raw_request is not None:
        request = _parse_value(raw_request, C.POD_GPU_REQUEST)
        if request > limit:
            raise LabelError(f"request {request} > limit {limit}")
    # Conscious fix of a reference quirk (pod.go:249-298 only checks the
    # pair when gpu_request is PRESENT): a whole-GPU limit with a
    # missing/unequal request would otherwise produce a multi-GPU pod
    # requesting 0 GPUs. Found by hypothesis (tests/test_properties.py).
    if limit > 1.0 and limit != request:
        raise LabelError(f"whole-GPU pods need limit == request "
                         f"(limit={limit}, request={request})")

    if limit == 0.0 and request == 0.0:
        return None  # regular pod (pod.go:303-305)

    memory = 0
    if raw_memory is not None:
        try:
            memory = int(raw_memory)
        except ValueError as e:
            raise LabelError(f"{C.POD_GPU_MEMORY} set error: {raw_memory!r}") from e
        if memory < 0:
            raise LabelError(f"{C.POD_GPU_MEMORY} negative: {memory}")

    raw_lease = labels.get(C.POD_LEASE_MS)
    if raw_lease is not None:
        try:
            lease = int(raw_lease)
        except ValueError as e:
            raise LabelError(f"{C.POD_LEASE_MS} set error: {raw_lease!r}") from e
        if not (0 < lease <= 10000):
            raise LabelError(f"{C.POD_LEASE_MS} out of range (0,10000]: "
                             f"{lease}")
        spec.lease_ms = lease

    spec.limit = limit
    spec.request = request
    spec.memory = memory
    spec.model = labels.get(C.POD_GPU_MODEL, "")
    return spec
