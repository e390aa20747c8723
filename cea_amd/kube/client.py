"""Minimal in-cluster Kubernetes REST client.

Role parity: the reference's client-go usage — BuildKubeClient
(/root/reference/pkg/gpu/nvidia/util/util.go:56-70), node condition updates
(health_checker.go:288-346), server-side-apply annotations
(version_visibility.go:67-86), pod listing / binding in the topology
scheduler (gke-topology-scheduler/schedule-daemon.py).  This image has no
kubernetes client package, so the stack carries its own small REST client
over `requests` (JSON content type; in-cluster service-account auth), plus
an in-memory FakeKubeClient mirroring client-go's fake.NewSimpleClientset
pattern used throughout the reference's tests (health_checker_test.go:235).
"""
from __future__ import annotations

import copy
import json
import logging
import os
import threading
from typing import Callable, Dict, List, Optional

log = logging.getLogger(__name__)

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class KubeError(RuntimeError):
    def __init__(self, status_code: int, message: str):
        super().__init__(f"kube api {status_code}: {message}")
        self.status_code = status_code


class KubeClient:
    """Tiny typed-enough wrapper over the REST API."""

    def __init__(
        self,
        base_url: Optional[str] = None,
        token: Optional[str] = None,
        ca_cert: Optional[str] = None,
    ):
        import requests

        if base_url is None:
            host = os.environ["KUBERNETES_SERVICE_HOST"]
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            base_url = f"https://{host}:{port}"
        self.base_url = base_url.rstrip("/")
        self.session = requests.Session()
        if token is None:
            token_path = os.path.join(SA_DIR, "token")
            if os.path.exists(token_path):
                with open(token_path) as f:
                    token = f.read().strip()
        if token:
            self.session.headers["Authorization"] = f"Bearer {token}"
        if ca_cert is None:
            ca_path = os.path.join(SA_DIR, "ca.crt")
            ca_cert = ca_path if os.path.exists(ca_path) else None
        self.session.verify = ca_cert if ca_cert else False

    def _req(self, method: str, path: str, body=None, params=None,
             content_type: str = "application/json"):
        url = self.base_url + path
        headers = {"Content-Type": content_type}
        json_body = None
        data = None
        if body is not None:
            if content_type.endswith("json"):
                json_body = body
            else:
                # apply-patch+yaml (server-side apply): JSON is a YAML
                # subset, so a JSON-serialized body is valid — passing the
                # dict as `data` would form-encode it.
                import json as _json

                data = _json.dumps(body)
        r = self.session.request(
            method, url, json=json_body, data=data,
            params=params, headers=headers, timeout=30,
        )
        if r.status_code >= 400:
            raise KubeError(r.status_code, r.text[:500])
        return r.json() if r.content else None

    # -- list/watch (informer substrate) -------------------------------------
    def list_raw(self, path: str, params: Optional[dict] = None) -> dict:
        """GET a collection, returning the WHOLE list object (metadata.
        resourceVersion included) — the watch bookmark an informer needs."""
        return self._req("GET", path, params=params)

    def watch(self, path: str, resource_version: str,
              params: Optional[dict] = None, timeout_s: int = 300):
        """Streamed watch: yields (type, object) tuples until the server
        closes the connection (callers loop + relist on 410 Gone).  Uses
        the chunked watch protocol: one JSON event per line."""
        import json as _json

        p = dict(params or {})
        p.update({
            "watch": "true",
            "resourceVersion": resource_version,
            "allowWatchBookmarks": "true",
            "timeoutSeconds": str(timeout_s),
        })
        r = self.session.get(self.base_url + path, params=p, stream=True,
                             timeout=timeout_s + 30)
        if r.status_code >= 400:
            raise KubeError(r.status_code, r.text[:500])
        try:
            for line in r.iter_lines():
                if not line:
                    continue
                ev = _json.loads(line)
                yield ev.get("type", ""), ev.get("object", {})
        finally:
            r.close()

    # -- nodes ---------------------------------------------------------------
    def get_node(self, name: str) -> dict:
        return self._req("GET", f"/api/v1/nodes/{name}")

    def list_nodes(self, label_selector: str = "") -> List[dict]:
        params = {"labelSelector": label_selector} if label_selector else None
        return self._req("GET", "/api/v1/nodes", params=params)["items"]

    def patch_node(self, name: str, patch: dict) -> dict:
        """Strategic-merge patch (labels/annotations)."""
        return self._req(
            "PATCH", f"/api/v1/nodes/{name}", body=patch,
            content_type="application/strategic-merge-patch+json",
        )

    def apply_node_annotations(self, name: str, annotations: Dict[str, str],
                               field_manager: str = "amd-gpu-device-plugin") -> dict:
        """Server-side-apply of metadata annotations (parity:
        version_visibility.go:67-86 with the same field-manager pattern)."""
        body = {
            "apiVersion": "v1",
            "kind": "Node",
            "metadata": {"name": name, "annotations": annotations},
        }
        return self._req(
            "PATCH",
            f"/api/v1/nodes/{name}?fieldManager={field_manager}&force=true",
            body=body,
            content_type="application/apply-patch+yaml",
        )

    def update_node_status(self, name: str, node: dict) -> dict:
        return self._req("PUT", f"/api/v1/nodes/{name}/status", body=node)

    # -- pods ----------------------------------------------------------------
    def list_pods(self, namespace: str = "", field_selector: str = "",
                  label_selector: str = "") -> List[dict]:
        path = (
            f"/api/v1/namespaces/{namespace}/pods" if namespace else "/api/v1/pods"
        )
        params = {}
        if field_selector:
            params["fieldSelector"] = field_selector
        if label_selector:
            params["labelSelector"] = label_selector
        return self._req("GET", path, params=params or None)["items"]

    def get_pod(self, namespace: str, name: str) -> dict:
        return self._req("GET", f"/api/v1/namespaces/{namespace}/pods/{name}")

    def replace_pod(self, namespace: str, name: str, pod: dict) -> dict:
        return self._req(
            "PUT", f"/api/v1/namespaces/{namespace}/pods/{name}", body=pod
        )

    def delete_pod(self, namespace: str, name: str) -> None:
        self._req("DELETE", f"/api/v1/namespaces/{namespace}/pods/{name}")

    def create_pod(self, namespace: str, pod: dict) -> dict:
        return self._req("POST", f"/api/v1/namespaces/{namespace}/pods", body=pod)

    # -- events --------------------------------------------------------------
    def create_event(self, namespace: str, event: dict) -> dict:
        return self._req(
            "POST", f"/api/v1/namespaces/{namespace}/events", body=event
        )


class FakeKubeClient:
    """In-memory fake with reactor injection, parity with the reference's
    fake.NewSimpleClientset + PrependReactor tests
    (health_checker_test.go:311-320)."""

    def __init__(self, nodes: Optional[List[dict]] = None,
                 pods: Optional[List[dict]] = None):
        self.nodes: Dict[str, dict] = {
            n["metadata"]["name"]: copy.deepcopy(n) for n in nodes or []
        }
        self.pods: Dict[tuple, dict] = {
            (p["metadata"].get("namespace", "default"), p["metadata"]["name"]):
                copy.deepcopy(p)
            for p in pods or []
        }
        self.events: List[dict] = []
        self.reactors: List[Callable] = []   # (verb, resource, obj) -> maybe raise
        self.lock = threading.Lock()
        self._rv = 1
        self._watch_queue: List[tuple] = []

    def _react(self, verb: str, resource: str, obj=None):
        for r in self.reactors:
            r(verb, resource, obj)

    def prepend_reactor(self, fn: Callable) -> None:
        self.reactors.insert(0, fn)

    # nodes
    def get_node(self, name: str) -> dict:
        with self.lock:
            self._react("get", "nodes", name)
            if name not in self.nodes:
                raise KubeError(404, f"node {name} not found")
            return copy.deepcopy(self.nodes[name])

    def list_nodes(self, label_selector: str = "") -> List[dict]:
        with self.lock:
            self._react("list", "nodes")
            out = list(copy.deepcopy(list(self.nodes.values())))
        if label_selector:
            want = dict(kv.split("=", 1) for kv in label_selector.split(","))
            out = [
                n for n in out
                if all(n["metadata"].get("labels", {}).get(k) == v
                       for k, v in want.items())
            ]
        return out

    def patch_node(self, name: str, patch: dict) -> dict:
        with self.lock:
            self._react("patch", "nodes", patch)
            node = self.nodes[name]
            meta = patch.get("metadata", {})
            for key in ("labels", "annotations"):
                if key in meta:
                    node.setdefault("metadata", {}).setdefault(key, {}).update(
                        {k: v for k, v in meta[key].items() if v is not None}
                    )
                    for k, v in meta[key].items():
                        if v is None:
                            node["metadata"][key].pop(k, None)
            return copy.deepcopy(node)

    def apply_node_annotations(self, name, annotations, field_manager="x"):
        return self.patch_node(name, {"metadata": {"annotations": annotations}})

    def update_node_status(self, name: str, node: dict) -> dict:
        with self.lock:
            self._react("update", "nodes/status", node)
            self.nodes[name] = copy.deepcopy(node)
            return copy.deepcopy(node)

    # pods
    def list_pods(self, namespace="", field_selector="", label_selector=""):
        with self.lock:
            self._react("list", "pods")
            pods = [
                copy.deepcopy(p) for (ns, _), p in self.pods.items()
                if not namespace or ns == namespace
            ]
        if field_selector:
            sels = dict(kv.split("=", 1) for kv in field_selector.split(","))
            phase = sels.get("status.phase")
            if phase:
                pods = [p for p in pods if p.get("status", {}).get("phase") == phase]
        return pods

    def get_pod(self, namespace, name):
        with self.lock:
            key = (namespace, name)
            if key not in self.pods:
                raise KubeError(404, f"pod {namespace}/{name} not found")
            return copy.deepcopy(self.pods[key])

    def replace_pod(self, namespace, name, pod):
        with self.lock:
            self._react("update", "pods", pod)
            self.pods[(namespace, name)] = copy.deepcopy(pod)
            return copy.deepcopy(pod)

    def delete_pod(self, namespace, name):
        with self.lock:
            self.pods.pop((namespace, name), None)

    def create_pod(self, namespace, pod):
        with self.lock:
            self.pods[(namespace, pod["metadata"]["name"])] = copy.deepcopy(pod)
            return copy.deepcopy(pod)

    def create_event(self, namespace: str, event: dict) -> dict:
        with self.lock:
            self._react("create", "events", event)
            self.events.append(copy.deepcopy(event))
            return event

    # -- list/watch (informer substrate) -------------------------------------
    def list_raw(self, path: str, params: Optional[dict] = None) -> dict:
        with self.lock:
            if path.endswith("/nodes"):
                items = [copy.deepcopy(n) for n in self.nodes.values()]
            else:
                items = [copy.deepcopy(p) for p in self.pods.values()]
        return {"metadata": {"resourceVersion": str(self._rv)},
                "items": items}

    def push_watch_event(self, ev_type: str, obj: dict) -> None:
        """Test hook: enqueue an event for watchers (and apply it to the
        store so list/watch stay consistent)."""
        with self.lock:
            self._rv += 1
            obj = copy.deepcopy(obj)
            obj.setdefault("metadata", {})["resourceVersion"] = str(self._rv)
            meta = obj["metadata"]
            if obj.get("kind") == "Node" or "nodeInfo" in obj.get("status", {}):
                store, key = self.nodes, meta.get("name", "")
            else:
                store, key = self.pods, (meta.get("namespace", "default"),
                                         meta.get("name", ""))
            if ev_type == "DELETED":
                store.pop(key, None)
            else:
                store[key] = obj
            self._watch_queue.append((ev_type, obj))

    def watch(self, path: str, resource_version: str,
              params: Optional[dict] = None, timeout_s: int = 300):
        import time as _time

        deadline = _time.monotonic() + min(timeout_s, 1.0)
        while _time.monotonic() < deadline:
            with self.lock:
                if self._watch_queue:
                    yield self._watch_queue.pop(0)
                    continue
            _time.sleep(0.01)


def build_kube_client() -> KubeClient:
    """Parity: BuildKubeClient (util.go:56-70) — in-cluster config."""
    return KubeClient()
