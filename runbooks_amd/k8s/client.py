"""Dynamic Kubernetes client interface + HTTPS implementation.

Role parity: reference internal/client/client.go:39-46 (client.Interface)
and controller-runtime's client used throughout internal/controller/.
Objects are plain dicts ("unstructured"); the api/ dataclasses convert
to/from them at the controller boundary.
"""
from __future__ import annotations

import json
import os
import ssl
import threading
import urllib.parse
import urllib.request
from typing import Any, Iterator, Optional


class NotFound(Exception):
    pass


class Conflict(Exception):
    pass


# apiVersion → (api path prefix, whether namespaced resources live under it)
_CORE = ""  # core/v1 → /api/v1


def _group_path(api_version: str) -> str:
    if api_version == "v1":
        return "/api/v1"
    return f"/apis/{api_version}"


# kind → plural for the kinds this platform touches (substratus CRDs +
# the built-ins its controllers create).
PLURALS = {
    "Model": "models", "Dataset": "datasets", "Server": "servers",
    "Notebook": "notebooks",
    "Pod": "pods", "Service": "services", "ServiceAccount": "serviceaccounts",
    "ConfigMap": "configmaps", "Secret": "secrets", "Namespace": "namespaces",
    "Job": "jobs", "Deployment": "deployments", "Event": "events",
    "CustomResourceDefinition": "customresourcedefinitions",
    "Lease": "leases",
}

CLUSTER_SCOPED = {"Namespace", "CustomResourceDefinition"}


def resource_path(api_version: str, kind: str, namespace: str = "",
                  name: str = "") -> str:
    plural = PLURALS[kind]
    p = _group_path(api_version)
    if kind not in CLUSTER_SCOPED and namespace:
        p += f"/namespaces/{namespace}"
    p += f"/{plural}"
    if name:
        p += f"/{name}"
    return p


class KubeClient:
    """The interface the reconcilers and CLI are written against."""

    def get(self, api_version: str, kind: str, namespace: str,
            name: str) -> Optional[dict]:
        raise NotImplementedError

    def list(self, api_version: str, kind: str, namespace: str = "",
             label_selector: str = "") -> list[dict]:
        raise NotImplementedError

    def create(self, obj: dict) -> dict:
        raise NotImplementedError

    def apply(self, obj: dict, field_manager: str = "runbooks-amd") -> dict:
        """Server-side apply (create-or-merge)."""
        raise NotImplementedError

    def update(self, obj: dict) -> dict:
        raise NotImplementedError

    def update_status(self, obj: dict) -> dict:
        raise NotImplementedError

    def patch(self, api_version: str, kind: str, namespace: str, name: str,
              patch: dict) -> dict:
        """Strategic-merge-style patch of the main resource."""
        raise NotImplementedError

    def delete(self, api_version: str, kind: str, namespace: str,
               name: str) -> bool:
        raise NotImplementedError

    def watch(self, api_version: str, kind: str, namespace: str = "",
              stop: Optional[threading.Event] = None) -> Iterator[dict]:
        """Yield {type: ADDED|MODIFIED|DELETED, object: {...}} events."""
        raise NotImplementedError

    # -- conveniences shared by both implementations
    def get_or_none(self, obj: dict) -> Optional[dict]:
        m = obj["metadata"]
        return self.get(obj["apiVersion"], obj["kind"],
                        m.get("namespace", "default"), m["name"])


class HTTPKubeClient(KubeClient):
    """Direct REST client for a real API server.

    In-cluster config from the standard service-account mount
    (/var/run/secrets/kubernetes.io/serviceaccount), or host/token/ca
    passed explicitly (tests point it at a local fake if ever needed).
    """

    SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"

    def __init__(self, host: str = "", token: str = "",
                 ca_file: str = "", insecure: bool = False):
        if not host:
            h = os.environ.get("KUBERNETES_SERVICE_HOST", "kubernetes.default")
            p = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            host = f"https://{h}:{p}"
        self.host = host.rstrip("/")
        if not token and os.path.exists(f"{self.SA_DIR}/token"):
            with open(f"{self.SA_DIR}/token") as f:
                token = f.read().strip()
        self.token = token
        if not ca_file and os.path.exists(f"{self.SA_DIR}/ca.crt"):
            ca_file = f"{self.SA_DIR}/ca.crt"
        if insecure:
            self._ctx = ssl._create_unverified_context()
        elif ca_file:
            self._ctx = ssl.create_default_context(cafile=ca_file)
        else:
            self._ctx = ssl.create_default_context()

    def _req(self, method: str, path: str, body: Optional[dict] = None,
             content_type: str = "application/json",
             query: Optional[dict] = None) -> Any:
        url = self.host + path
        if query:
            url += "?" + urllib.parse.urlencode(query)
        data = json.dumps(body).encode() if body is not None else None
        req = urllib.request.Request(url, data=data, method=method)
        req.add_header("Accept", "application/json")
        if data is not None:
            req.add_header("Content-Type", content_type)
        if self.token:
            req.add_header("Authorization", f"Bearer {self.token}")
        try:
            with urllib.request.urlopen(req, context=self._ctx) as resp:
                return json.loads(resp.read() or b"{}")
        except urllib.error.HTTPError as e:
            if e.code == 404:
                raise NotFound(path)
            if e.code == 409:
                raise Conflict(path)
            raise

    def _path_of(self, obj: dict, name: bool = True) -> str:
        m = obj["metadata"]
        return resource_path(obj["apiVersion"], obj["kind"],
                             m.get("namespace", "default"),
                             m["name"] if name else "")

    def get(self, api_version, kind, namespace, name):
        try:
            return self._req("GET", resource_path(api_version, kind,
                                                  namespace, name))
        except NotFound:
            return None

    def list(self, api_version, kind, namespace="", label_selector=""):
        q = {"labelSelector": label_selector} if label_selector else None
        out = self._req("GET", resource_path(api_version, kind, namespace),
                        query=q)
        return out.get("items", [])

    def create(self, obj):
        return self._req("POST", self._path_of(obj, name=False), obj)

    def apply(self, obj, field_manager="runbooks-amd"):
        return self._req(
            "PATCH", self._path_of(obj), obj,
            content_type="application/apply-patch+yaml",
            query={"fieldManager": field_manager, "force": "true"})

    def update(self, obj):
        return self._req("PUT", self._path_of(obj), obj)

    def update_status(self, obj):
        return self._req("PUT", self._path_of(obj) + "/status", obj)

    def patch(self, api_version, kind, namespace, name, patch):
        return self._req("PATCH",
                         resource_path(api_version, kind, namespace, name),
                         patch, content_type="application/merge-patch+json")

    def delete(self, api_version, kind, namespace, name):
        try:
            self._req("DELETE", resource_path(api_version, kind, namespace,
                                              name))
            return True
        except NotFound:
            return False

    def watch(self, api_version, kind, namespace="", stop=None):
        path = resource_path(api_version, kind, namespace)
        url = self.host + path + "?" + urllib.parse.urlencode({"watch": "1"})
        req = urllib.request.Request(url)
        req.add_header("Accept", "application/json")
        if self.token:
            req.add_header("Authorization", f"Bearer {self.token}")
        with urllib.request.urlopen(req, context=self._ctx) as resp:
            for line in resp:
                if stop is not None and stop.is_set():
                    return
                line = line.strip()
                if line:
                    yield json.loads(line)
