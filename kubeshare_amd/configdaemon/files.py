"""Per-GPU-UUID config files — the L2 <-> L1 contract.

Two files per physical GPU, named by its UUID (reference
pkg/config/query.go:70-105; consumed by the Gemini launcher via inotify,
launcher.py:89-98). gpu-schd (native/schd) parses the same format.

  <config dir>/<uuid>:
      n\n
      <namespace>/<name> <limit> <request> <memory>\n   (n lines)

  <podmanagerport dir>/<uuid>:
      n\n
      <namespace>/<name> <port>\n                       (n lines)

Writes are atomic (tmp file + os.replace) so a half-written file is never
observed by the inotify watcher — an improvement over the reference's
in-place Create+Write (query.go:82-93).
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Iterable


@dataclass(frozen=True)
class PodQuota:
    pod: str          # "namespace/name"
    limit: float
    request: float
    memory: int       # bytes; 0 = unlimited/default
    group: str = ""   # gang group: members are co-granted by gpu-schd
    lease_ms: int = 0  # latency class: lease-length override (q=NNN)

    def line(self) -> str:
        base = f"{self.pod} {_fmt(self.limit)} {_fmt(self.request)} {self.memory}"
        if self.group:
            base += f" {self.group}"
        if self.lease_ms:
            base += f" q={self.lease_ms}"
        return base + "\n"


@dataclass(frozen=True)
class PodPort:
    pod: str
    port: int

    def line(self) -> str:
        return f"{self.pod} {self.port}\n"


def _fmt(x: float) -> str:
    s = f"{x:.6f}".rstrip("0")
    return s + "0" if s.endswith(".") else s


def _atomic_write(path: str, content: str) -> None:
    tmp = path + ".tmp"
    with open(tmp, "w") as f:
        f.write(content)
        f.flush()
        os.fsync(f.fileno())
    os.replace(tmp, path)


def write_gpu_config(config_dir: str, uuid: str, quotas: Iterable[PodQuota]) -> str:
    quotas = list(quotas)
    path = os.path.join(config_dir, uuid)
    _atomic_write(path, f"{len(quotas)}\n" + "".join(q.line() for q in quotas))
    return path


def write_port_config(port_dir: str, uuid: str, ports: Iterable[PodPort]) -> str:
    ports = list(ports)
    path = os.path.join(port_dir, uuid)
    _atomic_write(path, f"{len(ports)}\n" + "".join(p.line() for p in ports))
    return path


def read_gpu_config(path: str) -> list[PodQuota]:
    with open(path) as f:
        lines = f.read().splitlines()
    if not lines:
        return []
    n = int(lines[0])
    out = []
    for line in lines[1:1 + n]:
        parts = line.split()
        pod, limit, request, memory = parts[:4]
        group, lease = "", 0
        for extra in parts[4:]:
            if extra.startswith("q="):
                lease = int(extra[2:])
            else:
                group = extra
        out.append(PodQuota(pod=pod, limit=float(limit), request=float(request),
                            memory=int(memory), group=group,
                            lease_ms=lease))
    return out


def read_port_config(path: str) -> list[PodPort]:
    with open(path) as f:
        lines = f.read().splitlines()
    if not lines:
        return []
    n = int(lines[0])
    out = []
    for line in lines[1:1 + n]:
        pod, port = line.split()
        out.append(PodPort(pod=pod, port=int(port)))
    return out


def zero_files(config_dir: str, port_dir: str) -> None:
    """Reset every per-UUID file to '0\\n' (reference query.go:115-138)."""
    for d in (config_dir, port_dir):
        for name in os.listdir(d):
            if name.endswith(".tmp"):
                continue
            _atomic_write(os.path.join(d, name), "0\n")
