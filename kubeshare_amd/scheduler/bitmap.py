"""Round-robin port pool for per-pod manager ports (reference
pkg/lib/bitmap rrbitmap.go:17-43; pool 50050..50561, node.go:13-15)."""
from __future__ import annotations

from ..utils import constants as C


class RRPortPool:
    def __init__(self, base: int = C.POD_MANAGER_PORT_START,
                 size: int = C.POD_MANAGER_PORT_POOL):
        self.base = base
        self.size = size
        self.used = [False] * size
        self._cursor = 0

    def available(self) -> bool:
        return not all(self.used)

    def allocate(self) -> int:
        """Next free port after the cursor (round-robin, so recently
        released ports are not immediately reused)."""
        for i in range(self.size):
            idx = (self._cursor + i) % self.size
            if not self.used[idx]:
                self.used[idx] = True
                self._cursor = (idx + 1) % self.size
                return self.base + idx
        raise RuntimeError("port pool exhausted")

    def mark(self, port: int):
        """Mark an externally-assigned port used (restart resync,
        reference pod.go:556-560)."""
        idx = port - self.base
        if 0 <= idx < self.size:
            self.used[idx] = True

    def release(self, port: int):
        idx = port - self.base
        if 0 <= idx < self.size:
            self.used[idx] = False

    def is_free(self, port: int) -> bool:
        idx = port - self.base
        return 0 <= idx < self.size and not self.used[idx]
