"""Minimal Kubernetes REST client — enough CoreV1 surface for the
KubeDriver, with no dependency on the `kubernetes` package.

The reference registers into kube-scheduler and talks to the API server
through client-go (cmd/kubeshare-scheduler/main.go:26-38); this build's
driver (kube.py) runs out-of-tree and only needs a handful of CoreV1
verbs: list nodes/pods, read/create/delete pods. Those are plain JSON
over HTTP, so a ~150-line stdlib client covers them — and the same
driver then runs unchanged against the in-process fake API server
(kubeshare_amd.testing.fake_apiserver) in CI, where the `kubernetes`
package and a kind cluster aren't available.

Objects are returned as `K8sObj` wrappers: attribute access in
snake_case maps onto the JSON camelCase (`pod.spec.node_name` ->
spec["nodeName"]), mutation writes through to the underlying dict, and
`to_payload()` serializes back — the same access pattern the official
client's models expose, so KubeDriver is client-agnostic.

Auth: in-cluster ServiceAccount bearer token when present; otherwise
anonymous (fake apiserver / `kubectl proxy`).
"""
from __future__ import annotations

import json
import os
import ssl
import urllib.parse
import urllib.request

_SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


def _camel(name: str) -> str:
    parts = name.split("_")
    return parts[0] + "".join(p.title() for p in parts[1:])


class K8sObj:
    """Attribute-access view over a JSON dict (snake_case -> camelCase),
    writing through to the underlying data."""

    def __init__(self, data: dict):
        object.__setattr__(self, "_data", data)

    def __getattr__(self, name: str):
        if name.startswith("_"):
            raise AttributeError(name)
        v = self._data.get(_camel(name))
        if isinstance(v, dict):
            return K8sObj(v)
        if isinstance(v, list):
            return [K8sObj(x) if isinstance(x, dict) else x for x in v]
        return v

    def __setattr__(self, name: str, value):
        key = _camel(name)
        if isinstance(value, K8sObj):
            value = value.to_payload()
        if isinstance(value, list):
            value = [x.to_payload() if isinstance(x, K8sObj) else x
                     for x in value]
        if value is None:
            self._data.pop(key, None)
        else:
            self._data[key] = value

    # dict protocol — labels/annotations/env maps read naturally
    # (`dict(x.items())`, `k in x`, `x or {}`)
    def get(self, name: str, default=None):
        return self._data.get(name, default)

    def items(self):
        return self._data.items()

    def keys(self):
        return self._data.keys()

    def values(self):
        return self._data.values()

    def __getitem__(self, key):
        return self._data[key]

    def __contains__(self, key):
        return key in self._data

    def __len__(self):
        return len(self._data)

    def __iter__(self):
        return iter(self._data)

    def to_payload(self) -> dict:
        return self._data

    def __repr__(self):
        return f"K8sObj({self._data.get('kind', '')} " \
               f"{self._data.get('metadata', {}).get('name', '')})"


class K8sList:
    def __init__(self, items: list):
        self.items = [K8sObj(x) if isinstance(x, dict) else x for x in items]


def _match_labels(obj: dict, selector: str) -> bool:
    labels = (obj.get("metadata") or {}).get("labels") or {}
    for term in filter(None, selector.split(",")):
        if "!=" in term:
            k, v = term.split("!=", 1)
            if labels.get(k.strip()) == v.strip():
                return False
        elif "=" in term:
            k, v = term.split("=", 1)
            if labels.get(k.strip()) != v.strip():
                return False
        else:
            if term.strip() not in labels:
                return False
    return True


def _match_fields(obj: dict, selector: str) -> bool:
    for term in filter(None, selector.split(",")):
        if "=" not in term:
            continue
        k, v = term.split("=", 1)
        cur = obj
        for part in k.strip().split("."):
            cur = (cur or {}).get(part)
        if (cur or "") != v.strip():
            return False
    return True


class RestCoreV1:
    """CoreV1Api-compatible subset over plain HTTP(S)."""

    def __init__(self, base_url: str = "", token: str = "",
                 verify_ssl: bool = True):
        if not base_url:
            host = os.environ.get("KUBERNETES_SERVICE_HOST")
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            if host:
                base_url = f"https://{host}:{port}"
            else:
                base_url = "http://127.0.0.1:8001"  # kubectl proxy default
        self.base_url = base_url.rstrip("/")
        if not token and os.path.exists(f"{_SA_DIR}/token"):
            with open(f"{_SA_DIR}/token") as f:
                token = f.read().strip()
        self.token = token
        self._ctx = None
        if self.base_url.startswith("https"):
            ca = f"{_SA_DIR}/ca.crt"
            if verify_ssl and os.path.exists(ca):
                self._ctx = ssl.create_default_context(cafile=ca)
            else:
                self._ctx = ssl._create_unverified_context()  # noqa: S323

    # ------------------------------------------------------------- transport
    def _req(self, method: str, path: str, params: dict | None = None,
             body: dict | None = None):
        url = self.base_url + path
        if params:
            url += "?" + urllib.parse.urlencode(
                {k: v for k, v in params.items() if v})
        data = json.dumps(body).encode() if body is not None else None
        req = urllib.request.Request(url, data=data, method=method)
        req.add_header("Accept", "application/json")
        if data is not None:
            req.add_header("Content-Type", "application/json")
        if self.token:
            req.add_header("Authorization", f"Bearer {self.token}")
        with urllib.request.urlopen(req, context=self._ctx, timeout=30) as r:
            return json.loads(r.read() or b"{}")

    # ----------------------------------------------------------------- verbs
    def list_node(self, label_selector: str = "") -> K8sList:
        out = self._req("GET", "/api/v1/nodes",
                        {"labelSelector": label_selector})
        return K8sList(out.get("items", []))

    def list_pod_for_all_namespaces(self, field_selector: str = "",
                                    label_selector: str = "") -> K8sList:
        out = self._req("GET", "/api/v1/pods",
                        {"fieldSelector": field_selector,
                         "labelSelector": label_selector})
        return K8sList(out.get("items", []))

    def list_namespaced_pod(self, namespace: str,
                            label_selector: str = "") -> K8sList:
        out = self._req("GET", f"/api/v1/namespaces/{namespace}/pods",
                        {"labelSelector": label_selector})
        return K8sList(out.get("items", []))

    def read_namespaced_pod(self, name: str, namespace: str) -> K8sObj:
        return K8sObj(self._req(
            "GET", f"/api/v1/namespaces/{namespace}/pods/{name}"))

    def create_namespaced_pod(self, namespace: str, body) -> K8sObj:
        if isinstance(body, K8sObj):
            body = body.to_payload()
        return K8sObj(self._req(
            "POST", f"/api/v1/namespaces/{namespace}/pods", body=body))

    def delete_namespaced_pod(self, name: str, namespace: str,
                              grace_period_seconds: int | None = None):
        params = {}
        if grace_period_seconds is not None:
            params["gracePeriodSeconds"] = str(grace_period_seconds)
        return self._req(
            "DELETE", f"/api/v1/namespaces/{namespace}/pods/{name}", params)

    def patch_namespaced_pod(self, name: str, namespace: str, body: dict):
        # strategic-merge patch is enough for annotation updates
        return self._merge_patch(
            f"/api/v1/namespaces/{namespace}/pods/{name}", body)

    def patch_node(self, name: str, body: dict):
        """Merge-patch a Node (the noded publishes its GPU inventory as
        the kubeshare.amd/gpus annotation this way)."""
        return self._merge_patch(f"/api/v1/nodes/{name}", body)

    def _merge_patch(self, path: str, body: dict):
        data = json.dumps(body).encode()
        req = urllib.request.Request(self.base_url + path, data=data,
                                     method="PATCH")
        req.add_header("Content-Type", "application/merge-patch+json")
        if self.token:
            req.add_header("Authorization", f"Bearer {self.token}")
        with urllib.request.urlopen(req, context=self._ctx, timeout=30) as r:
            return json.loads(r.read() or b"{}")


class RestCustomObjects:
    """CustomObjectsApi-compatible subset (SharePod controller)."""

    def __init__(self, core: RestCoreV1):
        self.core = core

    def list_cluster_custom_object(self, group, version, plural):
        return self.core._req("GET", f"/apis/{group}/{version}/{plural}")

    def create_namespaced_custom_object(self, group, version, namespace,
                                        plural, body):
        return self.core._req(
            "POST", f"/apis/{group}/{version}/namespaces/{namespace}/"
            f"{plural}", body=body)

    def get_namespaced_custom_object(self, group, version, namespace,
                                     plural, name):
        return self.core._req(
            "GET", f"/apis/{group}/{version}/namespaces/{namespace}/"
            f"{plural}/{name}")

    def patch_namespaced_custom_object_status(self, group, version,
                                              namespace, plural, name,
                                              body):
        return self.core._merge_patch(
            f"/apis/{group}/{version}/namespaces/{namespace}/{plural}/"
            f"{name}/status", body)
