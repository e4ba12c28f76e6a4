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
mechanism for parity (apply_placement) but preserve the original UID
linkage via an annotation so controllers can correlate; a
mutating-webhook flow (inject at create time from the
sharedgpu/gpu_uuid annotation) is the documented alternative
(SURVEY.md §7 phase 1.3) and shares the same env builder
(webhook.shared_pod_env) so both paths inject identically.

Client: the official `kubernetes` package when importable, else the
stdlib REST client (kubeclient.py) — which also runs against the
in-process fake API server in CI (tests/test_kube_e2e.py).

Gang semantics: pods parked at Permit are tracked with a deadline
(2 s x group headcount, reference scheduler.go:44,573); the run loop
expires them, unreserving the whole gang (reference Unreserve rejects
all waiting members, scheduler.go:534-549) — a partial gang never
leaks its reservations.
"""
from __future__ import annotations

import argparse
import time

from ..utils import constants as C
from ..webhook import SHARED_VOLUMES, shared_pod_env
from .inventory import FakeInventory
from .plugin import KubeShareScheduler, Placement
from .topology import TopologyConfig


def make_client():
    """Official client when available, stdlib REST client otherwise."""
    try:
        from kubernetes import client, config
        try:
            config.load_incluster_config()
        except Exception:  # noqa: BLE001
            config.load_kube_config()
        return client.CoreV1Api()
    except ImportError:
        from .kubeclient import RestCoreV1
        return RestCoreV1()


class KubeDriver:
    def __init__(self, topology: TopologyConfig, inventory=None, api=None):
        self.v1 = api if api is not None else make_client()
        self.sched = KubeShareScheduler(topology)
        self.inventory = inventory
        # gang members reserved but waiting at Permit: key -> (pod, placement)
        self.waiting_pods: dict = {}

    # ------------------------------------------------------------ nodes
    def sync_nodes(self):
        seen = set()
        for node in self.v1.list_node(
                label_selector="SharedGPU=true").items:
            name = node.metadata.name
            seen.add(name)
            if self.inventory is not None:
                by_model = self.inventory.by_model(name)
            else:
                # collector publishes inventory as node annotations
                # kubeshare.amd/gpus =
                #   "uuid,model,memory,index[,links=j:k:l];..."
                # (links = peer GPU indices with a live direct xGMI
                # link; absent -> assume the MI355X clique)
                raw = (node.metadata.annotations or {}).get(
                    "kubeshare.amd/gpus", "")
                by_model = {}
                for entry in filter(None, raw.split(";")):
                    fields = entry.split(",")
                    uuid, model, memory, index = fields[:4]
                    gpu = {"uuid": uuid, "memory": int(memory),
                           "index": int(index)}
                    for extra in fields[4:]:
                        if extra.startswith("links=") and \
                                extra != "links=":
                            gpu["xgmi_links"] = {
                                int(j): 1
                                for j in extra[len("links="):].split(":")
                                if j}
                    by_model.setdefault(model, []).append(gpu)
            ready = any(c.type == "Ready" and c.status == "True"
                        for c in (node.status.conditions or []))
            if by_model:
                self.sched.register_node(name, by_model, healthy=ready)
            else:
                self.sched.set_node_health(name, ready)
        # a topology node that lost its SharedGPU label or left the
        # cluster must stop receiving pods (reference updateNode label
        # handling, node.go:54-68)
        for n in self.sched.tree.all_nodes():
            if n not in seen:
                self.sched.set_node_health(n, False)

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
        self.expire_permits()
        all_pods = self.v1.list_pod_for_all_namespaces().items
        self.reconcile_deleted(all_pods)
        pods = [p for p in all_pods
                if p.spec.scheduler_name == C.SCHEDULER_NAME
                and not p.spec.node_name
                and (not p.status or p.status.phase in (None, "Pending"))]
        for p in sorted(pods, key=self._queue_key):
            if f"{p.metadata.namespace}/{p.metadata.name}" in \
                    self.waiting_pods:
                continue  # parked at Permit; released or expired, not rerun
            self.schedule_pod(p)

    def reconcile_deleted(self, all_pods):
        """Reclaim reservations of pods that left the cluster or
        completed (the reference's informer DeleteFunc / completed-pod
        path, pod.go:91-136 — without this a long-running driver leaks
        cell availability and manager ports until restart)."""
        live = {}
        for p in all_pods:
            phase = p.status.phase if p.status else ""
            if phase in ("Succeeded", "Failed"):
                continue
            key = f"{p.metadata.namespace}/{p.metadata.name}"
            ann = p.metadata.annotations or {}
            live[key] = (p.metadata.uid,
                         ann.get("kubeshare.amd/original-uid", ""),
                         p.spec.node_name or "")
        for key, spec in list(self.sched.pod_status.items()):
            if key in self.waiting_pods:
                continue  # parked here; expiry handles it
            cur = live.get(key)
            if cur is not None:
                uid, orig_uid, node = cur
                # same key, SAME instance: the shadow-recreated pod has
                # a new uid but carries the original-uid annotation
                if uid == spec.uid or orig_uid == spec.uid or \
                        not spec.uuids:
                    continue
                # same name but a NEW unbound instance (controller
                # recreated it): the old instance's reservations must
                # go before the new one is parsed over the spec
                if node:
                    continue  # bound namesake: resync owns this case
            ns, name = key.split("/", 1)
            self.sched.delete_pod(ns, name)
            self.event_key(key, "Reclaimed",
                           "pod gone/completed; resources reclaimed")

    def event_key(self, key, reason, message):
        print(f"[kubeshare-scheduler] {key}: {reason}: {message}",
              flush=True)

    def _queue_key(self, pod):
        """Reference QueueSort (scheduler.go:247-267): group priority
        desc, then creation time, then key."""
        prio = 0
        try:
            from ..utils.labels import parse_pod
            spec = parse_pod(pod.metadata.namespace, pod.metadata.name,
                             pod.metadata.labels or {})
            if spec is not None:
                prio = spec.priority
        except Exception:  # noqa: BLE001 — invalid labels sort last
            prio = -1000
        return (-prio, str(pod.metadata.creation_timestamp or ""),
                f"{pod.metadata.namespace}/{pod.metadata.name}")

    def expire_permits(self):
        """Reject gangs whose Permit wait expired; their members were
        never applied to the cluster, so only local reservations (cells,
        port) must be reclaimed."""
        for key in self.sched.expired_waiting():
            spec = self.sched.pod_status.get(key)
            entry = self.waiting_pods.pop(key, None)
            if spec is not None:
                self.sched.unreserve(spec)
            if entry is not None:
                self.event(entry[0], "GangTimeout",
                           "Permit wait expired; gang rejected")

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
            else:
                # REGULAR pod (no sharedgpu labels) that still named
                # this scheduler: inside kube-scheduler the framework's
                # default plugins would place it (reference
                # cmd/kubeshare-scheduler/main.go:26-38); out-of-tree we
                # must not strand it Pending — bind to the healthy node
                # with the fewest pods, no env injection.
                self.bind_regular(pod)
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
            self._apply_or_unreserve(pod, placement, spec.key)
            for key in release:
                w = self.waiting_pods.pop(key, None)
                if w is not None:
                    self._apply_or_unreserve(w[0], w[1], key)
        else:
            self.waiting_pods[spec.key] = (pod, placement)

    def _apply_or_unreserve(self, pod, placement, key):
        """Apply a placement; if the pod vanished (deleted mid-release)
        roll the reservation back so a later retry cannot double-charge
        the cells/port."""
        try:
            self.apply_placement(pod, placement)
        except Exception as e:  # noqa: BLE001
            self.event(pod, "ApplyFailed", str(e))
            spec = self.sched.pod_status.get(key)
            if spec is not None:
                self.sched.unreserve(spec)

    def bind_regular(self, pod):
        """Least-loaded-node placement for label-less pods that chose
        this scheduler anyway."""
        healthy = [n for n in self.sched.tree.all_nodes()
                   if any(c.healthy for c in self.sched.tree.node_cells[n])]
        if not healthy:
            self.event(pod, "Unschedulable", "no healthy node")
            return
        load = {n: 0 for n in healthy}
        for p in self.v1.list_pod_for_all_namespaces().items:
            if p.spec.node_name in load:
                load[p.spec.node_name] += 1
        node = min(sorted(healthy), key=lambda n: load[n])
        self.apply_placement(pod, Placement(
            node=node, uuids=[], cell_ids=[], gpu_indices=[],
            gpu_mem=0, manager_port=0))

    def placement_env(self, pod, placement: Placement) -> list:
        """Env entries (plain dicts — both clients serialize them) for
        the injected containers; identical to the webhook path."""
        if not placement.uuids:
            return []  # regular pod: no injection at all
        idx = ",".join(map(str, placement.gpu_indices))
        if placement.manager_port:
            return shared_pod_env(pod.metadata.namespace, pod.metadata.name,
                                  str(placement.manager_port),
                                  str(placement.gpu_mem), idx)
        return [{"name": C.ENV_INJECTED, "value": "1"},
                {"name": C.ENV_ROCR_VISIBLE_DEVICES, "value": idx}]

    def apply_placement(self, pod, placement: Placement):
        """Shadow-pod recreate with injected env + pinned node
        (reference pod.go:402-476)."""
        body = self.v1.read_namespaced_pod(pod.metadata.name,
                                           pod.metadata.namespace)
        original_uid = body.metadata.uid  # before nulling: `body` may be
        body.metadata.resource_version = None  # the same cached object
        body.metadata.uid = None
        ann = dict((body.metadata.annotations or {}).items())
        ann.update(placement.annotations)
        ann["kubeshare.amd/original-uid"] = original_uid
        body.metadata.annotations = ann
        body.spec.node_name = placement.node
        env = self.placement_env(pod, placement)
        containers = body.spec.containers
        for container in containers:
            container.env = (container.env or []) + env
            if placement.manager_port:
                container.volume_mounts = (container.volume_mounts or []) + [
                    {"name": "kubeshare-library",
                     "mountPath": C.LIBRARY_PATH, "readOnly": True},
                    {"name": "kubeshare-sock", "mountPath": C.SOCK_DIR},
                ]
        body.spec.containers = containers
        if placement.manager_port:
            body.spec.volumes = (body.spec.volumes or []) + \
                [dict(v) for v in SHARED_VOLUMES]
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
    ap.add_argument("--api-url", default="",
                    help="API server base URL (stdlib REST client)")
    args = ap.parse_args()
    topo = TopologyConfig.from_file(args.topology)
    inv = None
    if args.fake_nodes:
        inv = FakeInventory({f"node-{i}": {"gpus": 8}
                             for i in range(args.fake_nodes)})
    api = None
    if args.api_url:
        from .kubeclient import RestCoreV1
        api = RestCoreV1(args.api_url)
    KubeDriver(topo, inv, api=api).run(topology_path=args.topology)


if __name__ == "__main__":
    main()
