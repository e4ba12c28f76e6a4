"""Kubernetes driver for KubeShareScheduler — runs the plugin's cycle
against a real cluster as an out-of-tree scheduler.

The reference plugs into kube-scheduler's framework (cmd/
kubeshare-scheduler/main.go:26-38); this build keeps the extension-point
structure (plugin.py) but drives it from a small loop: watch Pending
pods with spec.schedulerName == kubeshare-scheduler, run
PreFilter/Filter/Score/Reserve/Permit, then APPLY the placement.

Applying a placement needs env injection, which k8s forbids on an
existing pod — the reference deletes and recreates a "shadow pod"
(pkg/scheduler/scheduler.go:515-528, pod.go:402-476). We keep that
mechanism for parity (recreate_with_injection) but preserve the
original UID linkage via an annotation so controllers can correlate;
a mutating-webhook flow (inject at create time from the
sharedgpu/gpu_uuid annotation) is the documented alternative
(SURVEY.md §7 phase 1.3).

Inventory comes from the collector's amdsmi export (or node labels),
injected as a provider — never from Prometheus inside the Filter hot
path (reference flaw, README.md:141).
"""
from __future__ import annotations

import argparse
import time

from ..utils import constants as C
from .inventory import FakeInventory
from .plugin import KubeShareScheduler, Placement
from .topology import TopologyConfig


class KubeDriver:
    def __init__(self, topology: TopologyConfig, inventory=None):
        from kubernetes import client, config, watch  # lazy: optional dep
        try:
            config.load_incluster_config()
        except Exception:  # noqa: BLE001
            config.load_kube_config()
        self.v1 = client.CoreV1Api()
        self.client = client
        self.watch = watch
        self.sched = KubeShareScheduler(topology)
        self.inventory = inventory
        # gang members reserved but waiting at Permit: key -> (pod, placement)
        self.waiting_pods: dict = {}

    # ------------------------------------------------------------ nodes
    def sync_nodes(self):
        for node in self.v1.list_node(
                label_selector="SharedGPU=true").items:
            name = node.metadata.name
            if self.inventory is not None:
                by_model = self.inventory.by_model(name)
            else:
                # collector publishes inventory as node annotations
                # kubeshare.amd/gpus = "uuid,model,memory,index;..."
                raw = (node.metadata.annotations or {}).get(
                    "kubeshare.amd/gpus", "")
                by_model = {}
                for entry in filter(None, raw.split(";")):
                    uuid, model, memory, index = entry.split(",")
                    by_model.setdefault(model, []).append(
                        {"uuid": uuid, "memory": int(memory),
                         "index": int(index)})
            ready = any(c.type == "Ready" and c.status == "True"
                        for c in (node.status.conditions or []))
            if by_model:
                self.sched.register_node(name, by_model, healthy=ready)
            else:
                self.sched.set_node_health(name, ready)

    # ------------------------------------------------------- resync
    def resync_bound(self):
        """Rebuild reservations after restart (reference bound-pod
        queue, pod.go:47-78, 528-617)."""
        for p in self.v1.list_pod_for_all_namespaces().items:
            if p.spec.scheduler_name != C.SCHEDULER_NAME:
                continue
            if not p.spec.node_name or (p.status and
                                        p.status.phase in ("Succeeded",
                                                           "Failed")):
                continue
            self.sched.resync_bound_pod(
                p.metadata.namespace, p.metadata.name,
                p.metadata.labels or {}, p.metadata.annotations or {},
                p.spec.node_name, uid=p.metadata.uid)

    # ---------------------------------------------------------- the loop
    def run_once(self):
        pods = [p for p in self.v1.list_pod_for_all_namespaces(
                    field_selector="status.phase=Pending").items
                if p.spec.scheduler_name == C.SCHEDULER_NAME
                and not p.spec.node_name]
        for p in sorted(pods,
                        key=lambda p: p.metadata.creation_timestamp or 0):
            self.schedule_pod(p)

    def schedule_pod(self, pod):
        ns, name = pod.metadata.namespace, pod.metadata.name
        labels = pod.metadata.labels or {}
        group = labels.get(C.POD_GROUP_NAME, "")
        in_group = None
        if group:
            in_group = len([
                q for q in self.v1.list_namespaced_pod(
                    ns, label_selector=f"{C.POD_GROUP_NAME}={group}").items])
        spec, err = self.sched.pre_filter(ns, name, labels,
                                          uid=pod.metadata.uid,
                                          all_pods_in_group=in_group)
        if spec is None:
            if err:
                self.event(pod, "PreFilterRejected", err)
            return
        feasible = []
        for node in self.sched.tree.all_nodes():
            ok, _ = self.sched.filter(spec, node)
            if ok:
                feasible.append(node)
        if not feasible:
            self.event(pod, "Unschedulable", "no feasible node")
            return
        scores = self.sched.normalize_scores(
            {n: self.sched.score(spec, n) for n in feasible})
        best = max(sorted(scores), key=lambda n: scores[n])
        placement = self.sched.reserve(spec, best)
        if placement is None:
            self.event(pod, "Unschedulable", "reserve failed")
            return
        decision, _, release = self.sched.permit(spec)
        if decision == "allow":
            self.apply_placement(pod, placement)
            for key in release:
                w = self.waiting_pods.pop(key, None)
                if w is not None:
                    self.apply_placement(w[0], w[1])
        else:
            self.waiting_pods[spec.key] = (pod, placement)

    def apply_placement(self, pod, placement: Placement):
        """Shadow-pod recreate with injected env + pinned node
        (reference pod.go:402-476)."""
        body = self.v1.read_namespaced_pod(pod.metadata.name,
                                           pod.metadata.namespace)
        original_uid = body.metadata.uid  # before nulling: `body` may be
        body.metadata.resource_version = None  # the same cached object
        body.metadata.uid = None
        body.metadata.annotations = dict(body.metadata.annotations or {})
        body.metadata.annotations.update(placement.annotations)
        body.metadata.annotations["kubeshare.amd/original-uid"] = \
            original_uid
        body.spec.node_name = placement.node
        env = [self.client.V1EnvVar(name=k, value=v)
               for k, v in placement.env.items()]
        for container in body.spec.containers:
            container.env = (container.env or []) + env
            if placement.manager_port:
                container.volume_mounts = (container.volume_mounts or []) + [
                    self.client.V1VolumeMount(
                        name="kubeshare-library",
                        mount_path=C.LIBRARY_PATH)]
        if placement.manager_port:
            body.spec.volumes = (body.spec.volumes or []) + [
                self.client.V1Volume(
                    name="kubeshare-library",
                    host_path=self.client.V1HostPathVolumeSource(
                        path=C.LIBRARY_PATH))]
        self.v1.delete_namespaced_pod(pod.metadata.name,
                                      pod.metadata.namespace,
                                      grace_period_seconds=0)
        self.v1.create_namespaced_pod(pod.metadata.namespace, body)

    def event(self, pod, reason, message):
        print(f"[kubeshare-scheduler] {pod.metadata.namespace}/"
              f"{pod.metadata.name}: {reason}: {message}", flush=True)

    def run(self, interval: float = 1.0, topology_path: str = ""):
        self.sync_nodes()
        self.resync_bound()
        last_gc = time.time()
        topo_mtime = self._mtime(topology_path)
        while True:
            try:
                self.run_once()
            except Exception as e:  # noqa: BLE001
                print(f"[kubeshare-scheduler] cycle error: {e}", flush=True)
            if time.time() - last_gc > C.POD_GROUP_GC_INTERVAL_SEC:
                self.sched.groups.gc()
                self.sync_nodes()
                last_gc = time.time()
                # restart-as-reload on topology change (the reference
                # exits and lets the Deployment restart it,
                # config.go:122-136; in-memory state rebuilds from the
                # bound-pod resync)
                if topology_path and self._mtime(topology_path) != topo_mtime:
                    print("[kubeshare-scheduler] topology changed; "
                          "exiting for restart-as-reload", flush=True)
                    raise SystemExit(0)
            time.sleep(interval)

    @staticmethod
    def _mtime(path: str):
        import os
        try:
            return os.stat(path).st_mtime if path else None
        except OSError:
            return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--topology",
                    default=C.CLUSTER_TOPOLOGY_FILE)
    ap.add_argument("--fake-nodes", type=int, default=0,
                    help="dev mode: N fake 8-GPU nodes instead of amdsmi")
    args = ap.parse_args()
    topo = TopologyConfig.from_file(args.topology)
    inv = None
    if args.fake_nodes:
        inv = FakeInventory({f"node-{i}": {"gpus": 8}
                             for i in range(args.fake_nodes)})
    KubeDriver(topo, inv).run(topology_path=args.topology)


if __name__ == "__main__":
    main()
