"""LocalGPUShare — stand up the full isolation chain for ONE GPU on the
local node: gpu-schd + one pod-mgr per pod + the env a shared-GPU pod
container would receive from the scheduler's Reserve step (reference
pod.go:402-476, launcher.py:13-31). Used by bench.py and the GPU tests;
in production the node daemon (kubeshare_amd.noded) plays this role.
"""
from __future__ import annotations

import math
import os
import socket
import subprocess
import tempfile
import time
from dataclasses import dataclass, field

from ..configdaemon import files as F
from ..utils import constants as C
from .client import query_stats

_REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
NATIVE_DIR = os.path.join(_REPO, "native")


def native_path(name: str) -> str:
    # in-tree build first (travels with the gpurun snapshot), then the
    # node-daemon install location
    p = os.path.join(NATIVE_DIR, name)
    if os.path.exists(p):
        return p
    p2 = os.path.join(C.LIBRARY_PATH, name)
    if os.path.exists(p2):
        return p2
    raise FileNotFoundError(
        f"{name} not built — run `make -C {NATIVE_DIR}` (or build())")


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _wait_port(port: int, timeout: float = 10.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            socket.create_connection(("127.0.0.1", port), timeout=0.3).close()
            return
        except OSError:
            time.sleep(0.05)
    raise TimeoutError(f"daemon on port {port} never came up")


def _wait_uds(path: str, timeout: float = 10.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if os.path.exists(path):
            try:
                s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
                s.settimeout(0.3)
                s.connect(path)
                s.close()
                return
            except OSError:
                pass
        time.sleep(0.05)
    raise TimeoutError(f"daemon at {path} never came up")


@dataclass
class PodHandle:
    name: str
    request: float
    limit: float
    memory: int
    manager_port: int = 0
    lease_ms: int = 0
    manager_uds: str = ""
    manager_proc: subprocess.Popen = None

    def env(self, gpu_index: int, base_env: dict | None = None) -> dict:
        """The env block the scheduler injects into the pod's container
        (ROCm-native equivalent of reference pod.go:445-457). UDS is the
        default transport; TCP stays populated as the fallback."""
        env = dict(base_env if base_env is not None else os.environ)
        env[C.ENV_ROCR_VISIBLE_DEVICES] = str(gpu_index)
        env[C.ENV_LD_PRELOAD] = native_path(C.HOOK_SO_NAME)
        if self.manager_uds:
            env[C.ENV_POD_MANAGER_UDS] = self.manager_uds
        env[C.ENV_POD_MANAGER_IP] = "127.0.0.1"
        env[C.ENV_POD_MANAGER_PORT] = str(self.manager_port)
        env[C.ENV_POD_NAME] = self.name
        env[C.ENV_GPU_MEM] = str(self.memory)
        env[C.ENV_REQUIRE_HOOK] = "1"
        return env


@dataclass
class LocalGPUShare:
    """One GPU's sharing stack: config file, gpu-schd, pod-mgrs."""

    gpu_index: int = 0
    uuid: str = ""
    base_quota_ms: float = C.BASE_QUOTA_MS
    min_quota_ms: float = C.MIN_QUOTA_MS
    window_ms: float = C.WINDOW_MS
    full_memory: int = C.MI355X_HBM_BYTES
    sched_port: int = 0
    workdir: str = ""
    pods: dict = field(default_factory=dict)
    _schd: subprocess.Popen = None

    def start(self):
        if not self.uuid:
            self.uuid = f"GPU-local-{self.gpu_index}"
        if not self.workdir:
            self.workdir = tempfile.mkdtemp(prefix="kubeshare-")
        self.config_dir = os.path.join(self.workdir, "config")
        self.port_dir = os.path.join(self.workdir, "podmanagerport")
        os.makedirs(self.config_dir, exist_ok=True)
        os.makedirs(self.port_dir, exist_ok=True)
        F.write_gpu_config(self.config_dir, self.uuid, [])
        log = os.path.join(self.workdir, f"gpu-schd-{self.gpu_index}.log")
        # deterministic UDS endpoint (keyed by workdir) — no bind races
        # when 8 ranks start their stacks concurrently, no hostNetwork
        self.sched_uds = os.path.join(self.workdir,
                                      f"schd-{self.gpu_index}.sock")
        self._schd = subprocess.Popen(
            [native_path("gpu-schd"), "-p", self.config_dir,
             "-f", self.uuid, "-U", self.sched_uds,
             "-q", str(self.base_quota_ms),
             "-m", str(self.min_quota_ms), "-w", str(self.window_ms),
             "-l", log, "-d", str(self.gpu_index)],
            stderr=subprocess.DEVNULL)
        _wait_uds(self.sched_uds, timeout=10.0)
        return self

    def add_pod(self, name: str, request: float, limit: float | None = None,
                memory: int = 0, lease_ms: int = 0) -> PodHandle:
        limit = limit if limit is not None else 1.0
        if memory <= 0:
            memory = math.floor(request * self.full_memory)
        h = PodHandle(name=name, request=request, limit=limit, memory=memory,
                      lease_ms=lease_ms, manager_port=0)
        h.manager_uds = os.path.join(
            self.workdir, "pm-" + name.replace("/", "_") + ".sock")
        h.manager_port = free_port()  # TCP fallback listener
        env = dict(os.environ)
        env.update({
            "SCHEDULER_UDS": self.sched_uds,
            C.ENV_POD_MANAGER_IP: "0.0.0.0",
            C.ENV_POD_MANAGER_PORT: str(h.manager_port),
            C.ENV_POD_MANAGER_UDS: h.manager_uds,
            C.ENV_POD_NAME: name,
            "POD_MANAGER_LOG": os.path.join(self.workdir, "pod-mgr.log"),
        })
        h.manager_proc = subprocess.Popen([native_path("pod-mgr")],
                                          env=env,
                                          stderr=subprocess.DEVNULL)
        _wait_uds(h.manager_uds, timeout=10.0)
        self.pods[name] = h
        self._rewrite_config()
        return h

    def remove_pod(self, name: str):
        h = self.pods.pop(name, None)
        if h and h.manager_proc:
            h.manager_proc.kill()
            h.manager_proc.wait()
        self._rewrite_config()

    def _rewrite_config(self):
        F.write_gpu_config(
            self.config_dir, self.uuid,
            [F.PodQuota(h.name, h.limit, h.request, h.memory,
                        lease_ms=h.lease_ms)
             for h in self.pods.values()])
        F.write_port_config(
            self.port_dir, self.uuid,
            [F.PodPort(h.name, h.manager_port) for h in self.pods.values()])

    def stats(self) -> dict:
        return query_stats(self.sched_uds, 0)

    def quota_error_pct(self) -> dict:
        """Per-pod |busy_share - request/sum(requests)| in % — the
        server-side quota-enforcement error (BASELINE.json metric)."""
        st = self.stats()
        if st.get("busy_ms", 0) <= 0:
            return {}
        reqs = {p: v["request"] for p, v in st["pods"].items()}
        total_req = sum(reqs.values()) or 1.0
        out = {}
        for p, v in st["pods"].items():
            want = reqs[p] / total_req
            out[p] = abs(v["busy_share"] - want) * 100.0
        return out

    def stop(self):
        for name in list(self.pods):
            self.remove_pod(name)
        if self._schd:
            self._schd.kill()
            self._schd.wait()
            self._schd = None
