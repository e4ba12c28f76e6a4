"""In-process fake Kubernetes API server (HTTP, stdlib only).

Serves the CoreV1 subset the KubeDriver uses — list/read/create/delete
pods, list nodes — over real HTTP with real JSON, so the full driver
loop (REST client -> selectors -> shadow-pod recreate -> webhook
mutation) is exercised end-to-end in CI where kind and the `kubernetes`
package don't exist. The reference has no such harness (its e2e tests
need a live lab cluster); this is the fixture layer that
makes BASELINE config #1 ("one pod gpu_request=0.5 schedules, plumbing
only, no GPU") an automated test instead of a manual install check.

Kubelet emulation (just enough): a pod created with spec.nodeName set
gets status.phase=Running and status.hostIP=<node's addr>, and its
downward-API env (`valueFrom.fieldRef status.hostIP`) is resolved the
way the real kubelet would.

Admission emulation: on every pod CREATE the registered mutator — by
default kubeshare_amd.webhook.admission_response, the very code a real
MutatingWebhookConfiguration would call over HTTPS — runs and its
JSONPatch is applied.
"""
from __future__ import annotations

import base64
import copy
import json
import threading
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

from ..scheduler.kubeclient import _match_fields, _match_labels


def apply_json_patch(doc: dict, patch: list) -> dict:
    """RFC-6902 subset: add / replace / remove, with '-' list append."""
    doc = copy.deepcopy(doc)
    for op in patch:
        parts = [p.replace("~1", "/").replace("~0", "~")
                 for p in op["path"].lstrip("/").split("/")]
        cur = doc
        for p in parts[:-1]:
            cur = cur[int(p)] if isinstance(cur, list) else cur.setdefault(p, {})
        last = parts[-1]
        kind = op["op"]
        if isinstance(cur, list):
            if kind == "add":
                if last == "-":
                    cur.append(op["value"])
                else:
                    cur.insert(int(last), op["value"])
            elif kind == "replace":
                cur[int(last)] = op["value"]
            elif kind == "remove":
                del cur[int(last)]
        else:
            if kind in ("add", "replace"):
                cur[last] = op["value"]
            elif kind == "remove":
                cur.pop(last, None)
    return doc


class FakeAPIServer:
    """State + HTTP front end. Thread-safe enough for test use (one
    lock around the stores)."""

    def __init__(self, mutator=None):
        # mutator(review_dict) -> AdmissionReview response dict;
        # default: the real webhook handler
        if mutator is None:
            from ..webhook import admission_response
            mutator = admission_response
        self.mutator = mutator
        self.nodes: dict[str, dict] = {}
        self.pods: dict[tuple, dict] = {}   # (ns, name) -> pod dict
        # custom resources: (group, plural, ns, name) -> object dict
        self.crs: dict[tuple, dict] = {}
        self.lock = threading.Lock()
        self._uid = 0
        self._now = 0
        self.deleted: list = []
        self.server: ThreadingHTTPServer | None = None

    # ------------------------------------------------------------- fixtures
    def add_node(self, name: str, labels: dict | None = None,
                 annotations: dict | None = None, ready: bool = True,
                 host_ip: str = ""):
        with self.lock:
            self.nodes[name] = {
                "kind": "Node",
                "metadata": {"name": name, "labels": labels or {},
                             "annotations": annotations or {}},
                "status": {
                    "conditions": [{"type": "Ready",
                                    "status": "True" if ready else "False"}],
                    "addresses": [{"type": "InternalIP",
                                   "address": host_ip or
                                   f"10.0.0.{len(self.nodes) + 1}"}],
                },
            }

    def submit_pod(self, namespace: str, name: str, labels: dict,
                   containers: list | None = None,
                   scheduler_name: str = "kubeshare-scheduler") -> dict:
        """Shortcut for tests: user `kubectl apply`s a pending pod."""
        pod = {
            "kind": "Pod",
            "metadata": {"namespace": namespace, "name": name,
                         "labels": dict(labels), "annotations": {}},
            "spec": {"schedulerName": scheduler_name,
                     "containers": containers or
                     [{"name": "main", "image": "rocm/pytorch"}]},
            "status": {"phase": "Pending"},
        }
        return self._create_pod(namespace, pod)

    # ------------------------------------------------------------ pod logic
    def _create_pod(self, namespace: str, pod: dict) -> dict:
        with self.lock:
            self._uid += 1
            self._now += 1
            meta = pod.setdefault("metadata", {})
            meta.setdefault("namespace", namespace)
            meta["uid"] = meta.get("uid") or f"uid-{self._uid}"
            meta["creationTimestamp"] = f"2026-01-01T00:00:{self._now:02d}Z"
            mutator = self.mutator
        if mutator is not None:
            review = {"apiVersion": "admission.k8s.io/v1",
                      "kind": "AdmissionReview",
                      "request": {"uid": "rev", "object": pod}}
            resp = mutator(review).get("response", {})
            if resp.get("patch"):
                patch = json.loads(base64.b64decode(resp["patch"]))
                pod = apply_json_patch(pod, patch)
        with self.lock:
            # kubelet emulation: node-pinned pod starts Running with the
            # downward API resolved
            if pod.get("spec", {}).get("nodeName"):
                node = self.nodes.get(pod["spec"]["nodeName"], {})
                addr = ""
                for a in node.get("status", {}).get("addresses", []):
                    if a.get("type") == "InternalIP":
                        addr = a.get("address", "")
                pod.setdefault("status", {})["phase"] = "Running"
                pod["status"]["hostIP"] = addr
                for c in pod["spec"].get("containers", []):
                    for e in c.get("env") or []:
                        ref = (e.get("valueFrom") or {}).get("fieldRef") or {}
                        if ref.get("fieldPath") == "status.hostIP":
                            e["value"] = addr
                            e.pop("valueFrom", None)
            else:
                # real apiserver: an unscheduled pod is phase Pending
                pod.setdefault("status", {}).setdefault("phase", "Pending")
            key = (pod["metadata"]["namespace"], pod["metadata"]["name"])
            self.pods[key] = pod
            return pod

    # ----------------------------------------------------------------- HTTP
    def start(self, port: int = 0) -> int:
        state = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):  # quiet
                pass

            def _send(self, obj, code=200):
                body = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def _route(self):
                u = urllib.parse.urlparse(self.path)
                q = dict(urllib.parse.parse_qsl(u.query))
                parts = [p for p in u.path.split("/") if p]
                return u, q, parts

            # /apis/{group}/{version}/... custom-resource routing:
            # returns (group, ns, plural, name, subresource) or None
            def _cr_route(self, parts):
                if len(parts) < 4 or parts[0] != "apis":
                    return None
                group = parts[1]
                rest = parts[3:]
                if rest and rest[0] == "namespaces" and len(rest) >= 3:
                    ns, plural = rest[1], rest[2]
                    name = rest[3] if len(rest) > 3 else None
                    sub = rest[4] if len(rest) > 4 else None
                    return group, ns, plural, name, sub
                return group, None, rest[0] if rest else None, None, None

            def do_GET(self):
                _, q, parts = self._route()
                lsel = q.get("labelSelector", "")
                fsel = q.get("fieldSelector", "")
                cr = self._cr_route(parts)
                if cr is not None:
                    group, ns, plural, name, _ = cr
                    with state.lock:
                        if name is not None:
                            obj = state.crs.get((group, plural, ns, name))
                            if obj is None:
                                return self._send({"kind": "Status",
                                                   "code": 404}, 404)
                            return self._send(obj)
                        items = [o for (g, p, n, _), o in state.crs.items()
                                 if g == group and p == plural
                                 and (ns is None or n == ns)]
                        return self._send({"kind": "List", "items": items})
                with state.lock:
                    if parts[:3] == ["api", "v1", "nodes"]:
                        items = [n for n in state.nodes.values()
                                 if _match_labels(n, lsel)]
                        return self._send({"kind": "NodeList", "items": items})
                    if parts[:3] == ["api", "v1", "pods"]:
                        items = [p for p in state.pods.values()
                                 if _match_labels(p, lsel)
                                 and _match_fields(p, fsel)]
                        return self._send({"kind": "PodList", "items": items})
                    # /api/v1/namespaces/<ns>/pods[/<name>]
                    if len(parts) >= 5 and parts[2] == "namespaces" and \
                            parts[4] == "pods":
                        ns = parts[3]
                        if len(parts) == 6:
                            pod = state.pods.get((ns, parts[5]))
                            if pod is None:
                                return self._send({"kind": "Status",
                                                   "code": 404}, 404)
                            return self._send(pod)
                        items = [p for (pns, _), p in state.pods.items()
                                 if pns == ns and _match_labels(p, lsel)
                                 and _match_fields(p, fsel)]
                        return self._send({"kind": "PodList", "items": items})
                self._send({"kind": "Status", "code": 404}, 404)

            def do_POST(self):
                _, _, parts = self._route()
                n = int(self.headers.get("Content-Length", 0))
                body = json.loads(self.rfile.read(n) or b"{}")
                cr = self._cr_route(parts)
                if cr is not None and cr[1] is not None:
                    group, ns, plural, _, _ = cr
                    with state.lock:
                        state._uid += 1
                        meta = body.setdefault("metadata", {})
                        meta.setdefault("namespace", ns)
                        meta.setdefault("uid", f"uid-{state._uid}")
                        state.crs[(group, plural, ns,
                                   meta.get("name"))] = body
                    return self._send(body, 201)
                if len(parts) == 5 and parts[4] == "pods":
                    ns = parts[3]
                    created = state._create_pod(ns, body)
                    return self._send(created, 201)
                self._send({"kind": "Status", "code": 404}, 404)

            def do_PATCH(self):
                _, _, parts = self._route()
                n = int(self.headers.get("Content-Length", 0))
                body = json.loads(self.rfile.read(n) or b"{}")
                # /api/v1/namespaces/<ns>/pods/<name> merge patch
                if len(parts) == 6 and parts[2] == "namespaces" and \
                        parts[4] == "pods":
                    with state.lock:
                        pod = state.pods.get((parts[3], parts[5]))
                        if pod is None:
                            return self._send({"kind": "Status",
                                               "code": 404}, 404)
                        meta = body.get("metadata", {})
                        pod.setdefault("metadata", {}).setdefault(
                            "annotations", {}).update(
                            meta.get("annotations") or {})
                        pod["metadata"].setdefault("labels", {}).update(
                            meta.get("labels") or {})
                        return self._send(pod)
                # /api/v1/nodes/<name> merge patch (inventory annotation)
                if parts[:3] == ["api", "v1", "nodes"] and len(parts) == 4:
                    with state.lock:
                        node = state.nodes.get(parts[3])
                        if node is None:
                            return self._send({"kind": "Status",
                                               "code": 404}, 404)
                        meta = body.get("metadata", {})
                        node.setdefault("metadata", {}).setdefault(
                            "annotations", {}).update(
                            meta.get("annotations") or {})
                        node["metadata"].setdefault("labels", {}).update(
                            meta.get("labels") or {})
                        return self._send(node)
                cr = self._cr_route(parts)
                if cr is not None and cr[3] is not None:
                    group, ns, plural, name, sub = cr
                    with state.lock:
                        obj = state.crs.get((group, plural, ns, name))
                        if obj is None:
                            return self._send({"kind": "Status",
                                               "code": 404}, 404)
                        # merge-patch: top-level keys replace
                        for k, v in body.items():
                            if isinstance(v, dict) and \
                                    isinstance(obj.get(k), dict):
                                obj[k].update(v)
                            else:
                                obj[k] = v
                        return self._send(obj)
                self._send({"kind": "Status", "code": 404}, 404)

            def do_DELETE(self):
                _, _, parts = self._route()
                if len(parts) == 6 and parts[4] == "pods":
                    key = (parts[3], parts[5])
                    with state.lock:
                        pod = state.pods.pop(key, None)
                        if pod is not None:
                            state.deleted.append(key)
                    return self._send({"kind": "Status",
                                       "status": "Success"})
                self._send({"kind": "Status", "code": 404}, 404)

        self.server = ThreadingHTTPServer(("127.0.0.1", port), Handler)
        t = threading.Thread(target=self.server.serve_forever, daemon=True)
        t.start()
        return self.server.server_address[1]

    def stop(self):
        if self.server is not None:
            self.server.shutdown()
            self.server.server_close()
            self.server = None
