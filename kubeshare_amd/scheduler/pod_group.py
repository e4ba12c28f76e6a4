"""Gang (co-)scheduling pod groups (reference pkg/scheduler/pod_group.go).

minAvailable = floor(headcount * threshold + 0.5) (pod_group.go:114) or
the direct sharedgpu/min_available label; groups expire after 600 s and
are GC'd every 30 s (scheduler.go:44-47, pod_group.go:119-129).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field


@dataclass
class PodGroupInfo:
    key: str                     # "<namespace>/<group name>"
    name: str
    priority: int = 0
    min_available: int = 0
    timestamp: float = field(default_factory=time.time)
    deleted: bool = False
    last_seen: float = field(default_factory=time.time)


class PodGroupRegistry:
    def __init__(self, expiration_sec: float = 600.0):
        self.groups: dict[str, PodGroupInfo] = {}
        self.expiration = expiration_sec

    @staticmethod
    def key_for(namespace: str, group: str) -> str:
        return f"{namespace}/{group}"

    def get_or_create(self, namespace: str, group: str, priority: int,
                      min_available: int) -> PodGroupInfo:
        key = self.key_for(namespace, group)
        info = self.groups.get(key)
        if info is None or info.deleted:
            info = PodGroupInfo(key=key, name=group, priority=priority,
                                min_available=min_available)
            self.groups[key] = info
        info.last_seen = time.time()
        # pods of one group must agree on these (PreFilter sanity,
        # reference scheduler.go:300-314)
        return info

    def gc(self, now: float | None = None):
        now = now if now is not None else time.time()
        for key in list(self.groups):
            if now - self.groups[key].last_seen > self.expiration:
                del self.groups[key]

    def remove(self, namespace: str, group: str):
        self.groups.pop(self.key_for(namespace, group), None)
