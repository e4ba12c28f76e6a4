"""Python-side client for the token protocol (native/common/protocol.hpp)
— used by tests, tooling and bench.py to query gpu-schd STATS and to
model pods in CPU-only tests."""
from __future__ import annotations

import json
import socket
import time


class TokenClient:
    def __init__(self, host: str, port: int, pod: str, timeout: float = 30.0):
        self._addr = (host, port)
        self._pod = pod
        self._timeout = timeout
        self._sock: socket.socket | None = None
        self._file = None

    def _ensure(self):
        if self._sock is None:
            host = self._addr[0]
            if host.startswith("/"):  # UDS path (default pod transport)
                s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
                s.settimeout(self._timeout)
                s.connect(host)
                self._sock = s
            else:
                self._sock = socket.create_connection(
                    self._addr, timeout=self._timeout)
                self._sock.setsockopt(socket.IPPROTO_TCP,
                                      socket.TCP_NODELAY, 1)
            self._file = self._sock.makefile("r")
        return self._sock

    def acquire(self, hint_ms: float = 0.0) -> float:
        """Blocks until GRANT; returns quota_ms."""
        s = self._ensure()
        s.sendall(f"REQ {self._pod} {hint_ms:.3f}\n".encode())
        old = self._sock.gettimeout()
        self._sock.settimeout(None)  # GRANT delay IS the throttle
        try:
            while True:
                line = self._file.readline()
                if not line:
                    raise ConnectionError("scheduler closed connection")
                if line.startswith("GRANT"):
                    return float(line.split()[1])
                if line.startswith("OK"):
                    continue
                raise RuntimeError(f"unexpected reply: {line!r}")
        finally:
            self._sock.settimeout(old)

    def release(self, used_ms: float) -> None:
        s = self._ensure()
        s.sendall(f"RET {self._pod} {used_ms:.3f}\n".encode())
        # OK consumed lazily by the next acquire()

    def close(self):
        if self._sock is not None:
            self._sock.close()
            self._sock = None


def query_stats(host: str, port: int = 0, timeout: float = 10.0) -> dict:
    """One STATS round-trip; `host` may be a UDS path ('/...')."""
    if host.startswith("/"):
        s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        s.settimeout(timeout)
        s.connect(host)
    else:
        s = socket.create_connection((host, port), timeout=timeout)
    try:
        s.sendall(b"STATS\n")
        f = s.makefile("r")
        deadline = time.time() + timeout
        while time.time() < deadline:
            line = f.readline()
            if not line:
                break
            if line.startswith("{"):
                return json.loads(line)
        raise TimeoutError("no STATS reply")
    finally:
        s.close()
