"""Node daemon launcher — supervises the per-GPU isolation stack.

MI355X-native replacement for the reference's gemini-scheduler container
(docker/kubeshare-gemini-scheduler/launcher-multigpus.sh +
launcher.py): enumerates GPUs (amdsmi -> rocm-smi -> torch -> fake),
seeds the per-UUID config files, starts one gpu-schd per GPU on port
BASE+index (launcher-multigpus.sh:21-42), watches the podmanagerport
directory and spawns/kills one pod-mgr per sharing pod as files change
(reference launcher.py:34-98 used inotify IN_CLOSE_WRITE; we poll mtimes
— the files are rewritten atomically by configdaemon, so a poll never
observes a torn file).

Runs as a plain process (DaemonSet container in production, subprocess
in tests):  python -m kubeshare_amd.noded.launcher --workdir /kubeshare
"""
from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys
import time

from ..configdaemon import files as F
from ..isolation.local import native_path
from ..utils import constants as C


def log(msg):
    print(f"[noded] {msg}", file=sys.stderr, flush=True)


def enumerate_gpus(source: str = "auto"):
    """Returns list[GPUInfo-like dicts]: uuid, model, memory, index."""
    if source in ("auto", "amdsmi"):
        try:
            from ..scheduler.inventory import AmdSmiInventory
            gpus = AmdSmiInventory().local_gpus()
            if gpus:
                return [g.__dict__ for g in gpus]
        except Exception as e:  # noqa: BLE001
            if source == "amdsmi":
                raise
            log(f"amdsmi unavailable ({e})")
    if source in ("auto", "torch"):
        try:
            from ..scheduler.inventory import TorchInventory
            gpus = TorchInventory().local_gpus()
            if gpus:
                return [g.__dict__ for g in gpus]
        except Exception as e:  # noqa: BLE001
            if source == "torch":
                raise
            log(f"torch inventory unavailable ({e})")
    return []


class PodManagerSupervisor:
    """Diffs one GPU's podmanagerport file against running pod-mgr
    processes (reference launcher.py:34-67)."""

    def __init__(self, uuid: str, sched_port: int, log_dir: str,
                 sock_dir: str = ""):
        self.uuid = uuid
        self.sched_port = sched_port
        self.log_dir = log_dir
        self.sock_dir = sock_dir
        self.procs: dict[str, subprocess.Popen] = {}   # pod -> proc
        self.ports: dict[str, int] = {}

    def reconcile(self, entries: list):
        want = {e.pod: e.port for e in entries}
        for pod in list(self.procs):
            if pod not in want or self.ports.get(pod) != want[pod] or \
                    self.procs[pod].poll() is not None:
                self._stop(pod)
        for pod, port in want.items():
            if pod not in self.procs:
                self._start(pod, port)

    def _start(self, pod: str, port: int):
        env = dict(os.environ)
        env.update({
            C.ENV_SCHEDULER_IP: "127.0.0.1",
            C.ENV_SCHEDULER_PORT: str(self.sched_port),
            C.ENV_POD_MANAGER_IP: "0.0.0.0",
            C.ENV_POD_MANAGER_PORT: str(port),
            C.ENV_POD_NAME: pod,
            "POD_MANAGER_LOG": os.path.join(self.log_dir, "pod-manager.log"),
        })
        if self.sock_dir:
            # default transport: per-pod UDS through the shared hostPath
            # (pod-mgr listens on BOTH; TCP stays as fallback)
            env[C.ENV_POD_MANAGER_UDS] = C.pod_manager_uds(
                port, self.sock_dir)
        self.procs[pod] = subprocess.Popen([native_path("pod-mgr")], env=env,
                                           stderr=subprocess.DEVNULL)
        self.ports[pod] = port
        log(f"pod-mgr started pod={pod} port={port} gpu={self.uuid}")

    def _stop(self, pod: str):
        proc = self.procs.pop(pod, None)
        self.ports.pop(pod, None)
        if proc is not None and proc.poll() is None:
            proc.kill()
            proc.wait()
        log(f"pod-mgr stopped pod={pod} gpu={self.uuid}")

    def stop_all(self):
        for pod in list(self.procs):
            self._stop(pod)


class NodeDaemon:
    def __init__(self, workdir: str, base_port: int = C.BASE_SCHED_PORT,
                 base_quota: float = C.BASE_QUOTA_MS,
                 min_quota: float = C.MIN_QUOTA_MS,
                 window: float = C.WINDOW_MS,
                 inventory_source: str = "auto",
                 gpus: list | None = None):
        self.workdir = workdir
        self.config_dir = os.path.join(workdir, "scheduler", "config")
        self.port_dir = os.path.join(workdir, "scheduler", "podmanagerport")
        self.log_dir = os.path.join(workdir, "log")
        self.sock_dir = os.path.join(workdir, "sock")
        for d in (self.config_dir, self.port_dir, self.log_dir,
                  self.sock_dir):
            os.makedirs(d, exist_ok=True)
        self.base_port = base_port
        self.knobs = (base_quota, min_quota, window)
        self.gpus = gpus if gpus is not None else \
            enumerate_gpus(inventory_source)
        self.schd: dict[str, subprocess.Popen] = {}
        self.sup: dict[str, PodManagerSupervisor] = {}
        self._mtimes: dict[str, float] = {}
        self.running = True

    def start(self):
        self.gpu_index = {g["uuid"]: g["index"] for g in self.gpus}
        for gpu in self.gpus:
            uuid, idx = gpu["uuid"], gpu["index"]
            port = self.base_port + idx
            cfg = os.path.join(self.config_dir, uuid)
            if not os.path.exists(cfg):
                F.write_gpu_config(self.config_dir, uuid, [])
            pf = os.path.join(self.port_dir, uuid)
            if not os.path.exists(pf):
                F.write_port_config(self.port_dir, uuid, [])
            self._spawn_schd(uuid, port, idx)
            self.sup[uuid] = PodManagerSupervisor(uuid, port, self.log_dir,
                                                  self.sock_dir)
        return self

    def _spawn_schd(self, uuid: str, port: int, gpu_index: int = -1):
        q, m, w = self.knobs
        cmd = [native_path("gpu-schd"), "-p", self.config_dir, "-f", uuid,
               "-P", str(port), "-q", str(q), "-m", str(m), "-w", str(w),
               "-l", os.path.join(self.log_dir, "gpu-schd.log")]
        if gpu_index >= 0:
            # enable the server-side busy sampler (leases charged
            # sampled GPU time; falls back to wall/RET off-GPU)
            cmd += ["-d", str(gpu_index)]
        self.schd[uuid] = subprocess.Popen(cmd, stderr=subprocess.DEVNULL)
        log(f"gpu-schd started gpu={uuid} port={port}")

    def poll_once(self):
        # a crashed gpu-schd is restarted (SO_REUSEADDR: the port is
        # rebindable immediately); hook clients reconnect on their next
        # renewal and pod-mgrs are respawned below if they exited with it
        for uuid, proc in list(self.schd.items()):
            if proc.poll() is not None:
                log(f"gpu-schd for {uuid} died (rc={proc.returncode}); "
                    f"restarting")
                self._spawn_schd(uuid, self.sup[uuid].sched_port,
                                 self.gpu_index.get(uuid, -1))
        for uuid, sup in self.sup.items():
            path = os.path.join(self.port_dir, uuid)
            try:
                mtime = os.stat(path).st_mtime
            except FileNotFoundError:
                continue
            if self._mtimes.get(uuid) == mtime:
                # still reconcile dead pod-mgr processes
                sup.reconcile([F.PodPort(p, sup.ports[p])
                               for p in sup.procs])
                continue
            self._mtimes[uuid] = mtime
            try:
                entries = F.read_port_config(path)
            except (ValueError, OSError) as e:
                log(f"bad port file {path}: {e}")
                continue
            sup.reconcile(entries)

    def publish_inventory(self, node_name: str = "", api=None) -> str:
        """Patch this Node's kubeshare.amd/gpus annotation (inventory +
        xGMI link graph) so the cluster scheduler can consume it without
        a Prometheus round-trip. Also labels the node SharedGPU=true."""
        from ..scheduler.inventory import GPUInfo, format_node_annotation
        infos = [g if isinstance(g, GPUInfo) else GPUInfo(**g)
                 for g in self.gpus]
        ann = format_node_annotation(infos)
        node = node_name or os.uname().nodename
        if api is None:
            from ..scheduler.kube import make_client
            api = make_client()
        patch = {"metadata": {"annotations": {"kubeshare.amd/gpus": ann},
                              "labels": {"SharedGPU": "true"}}}
        # both the official CoreV1Api and the stdlib RestCoreV1 expose
        # patch_node(name, body)
        api.patch_node(node, patch)
        return ann

    def serve_metrics(self, port: int, node_name: str = ""):
        """Export every local gpu-schd's STATS as Prometheus metrics."""
        from .metrics import serve
        endpoints = {uuid: ("127.0.0.1", sup.sched_port)
                     for uuid, sup in self.sup.items()}
        return serve(endpoints, node_name or os.uname().nodename, port)

    def run(self, interval: float = 0.5):
        signal.signal(signal.SIGTERM, lambda *_: self.stop())
        while self.running:
            self.poll_once()
            time.sleep(interval)

    def stop(self):
        self.running = False
        for sup in self.sup.values():
            sup.stop_all()
        for proc in self.schd.values():
            if proc.poll() is None:
                proc.kill()
                proc.wait()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workdir", default=C.KUBESHARE_ROOT)
    ap.add_argument("--base-port", type=int, default=C.BASE_SCHED_PORT)
    ap.add_argument("--base-quota", type=float, default=C.BASE_QUOTA_MS)
    ap.add_argument("--min-quota", type=float, default=C.MIN_QUOTA_MS)
    ap.add_argument("--window", type=float, default=C.WINDOW_MS)
    ap.add_argument("--inventory", default="auto",
                    choices=["auto", "amdsmi", "torch"])
    ap.add_argument("--metrics-port", type=int, default=0,
                    help="serve per-pod quota-enforcement metrics")
    ap.add_argument("--publish-inventory", action="store_true",
                    help="patch this Node's kubeshare.amd/gpus "
                         "annotation + SharedGPU label at startup")
    ap.add_argument("--node-name", default=os.environ.get("NODE_NAME", ""))
    args = ap.parse_args()
    daemon = NodeDaemon(args.workdir, args.base_port, args.base_quota,
                        args.min_quota, args.window, args.inventory)
    if not daemon.gpus:
        log("no GPUs found; exiting")
        return 1
    daemon.start()
    if args.publish_inventory:
        try:
            daemon.publish_inventory(args.node_name)
        except Exception as e:  # noqa: BLE001 — annotation is best-effort
            log(f"inventory publish failed: {e}")
    if args.metrics_port:
        daemon.serve_metrics(args.metrics_port)
    try:
        daemon.run()
    finally:
        daemon.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
