"""In-memory Kubernetes API server semantics for integration tests.

Plays the role envtest plays for the reference (real API server, no kubelet:
reference internal/controller/main_test.go:46-191, Makefile:96-99). Pods and
Jobs are stored but never run — tests patch their status by hand exactly like
the reference's fakeJobComplete / fakePodReady helpers
(reference main_test.go:245-265).
"""
from __future__ import annotations

import copy
import queue
import threading
from typing import Iterator

from .client import Conflict, KubeClient, NotFound


def _key(api_version: str, kind: str, ns: str, name: str):
    return (api_version, kind, ns or "", name)


def _deep_merge(dst: dict, src: dict) -> dict:
    """Merge-patch semantics: dicts merge recursively, None deletes,
    everything else replaces."""
    for k, v in src.items():
        if v is None:
            dst.pop(k, None)
        elif isinstance(v, dict) and isinstance(dst.get(k), dict):
            _deep_merge(dst[k], v)
        else:
            dst[k] = copy.deepcopy(v)
    return dst


class MemoryKubeClient(KubeClient):
    def __init__(self):
        self._store: dict[tuple, dict] = {}
        self._rv = 0
        self._lock = threading.RLock()
        self._watchers: list[tuple[tuple, queue.Queue]] = []

    # -- internals ---------------------------------------------------------
    def _bump(self, obj: dict, *, new: bool) -> dict:
        self._rv += 1
        m = obj.setdefault("metadata", {})
        m.setdefault("namespace", "default")
        m["resourceVersion"] = str(self._rv)
        if new:
            m.setdefault("uid", f"uid-{self._rv}")
            m.setdefault("generation", 1)
        return obj

    def _emit(self, type_: str, obj: dict) -> None:
        sel = (obj["apiVersion"], obj["kind"],
               obj["metadata"].get("namespace", "default"))
        for (want, q) in self._watchers:
            if want[0] == sel[0] and want[1] == sel[1] and \
                    (not want[2] or want[2] == sel[2]):
                q.put({"type": type_, "object": copy.deepcopy(obj)})

    def _k(self, obj: dict):
        m = obj["metadata"]
        return _key(obj["apiVersion"], obj["kind"],
                    m.get("namespace", "default"), m["name"])

    # -- KubeClient --------------------------------------------------------
    def get(self, api_version, kind, namespace, name):
        with self._lock:
            o = self._store.get(_key(api_version, kind, namespace, name))
            return copy.deepcopy(o) if o else None

    def list(self, api_version, kind, namespace="", label_selector=""):
        want_labels = {}
        if label_selector:
            for part in label_selector.split(","):
                k, _, v = part.partition("=")
                want_labels[k] = v
        out = []
        with self._lock:
            for (av, kd, ns, _), o in self._store.items():
                if av != api_version or kd != kind:
                    continue
                if namespace and ns != namespace:
                    continue
                labels = (o["metadata"].get("labels") or {})
                if all(labels.get(k) == v for k, v in want_labels.items()):
                    out.append(copy.deepcopy(o))
        return out

    def create(self, obj):
        obj = copy.deepcopy(obj)
        with self._lock:
            k = self._k(obj)
            if k in self._store:
                raise Conflict(str(k))
            self._bump(obj, new=True)
            self._store[k] = obj
            self._emit("ADDED", obj)
            return copy.deepcopy(obj)

    def apply(self, obj, field_manager="runbooks-amd"):
        obj = copy.deepcopy(obj)
        with self._lock:
            k = self._k(obj)
            cur = self._store.get(k)
            if cur is None:
                return self.create(obj)
            merged = copy.deepcopy(cur)
            spec_changed = "spec" in obj and obj["spec"] != cur.get("spec")
            patch = {kk: vv for kk, vv in obj.items() if kk != "status"}
            _deep_merge(merged, patch)
            if spec_changed:
                merged["metadata"]["generation"] = \
                    int(merged["metadata"].get("generation", 1)) + 1
            self._bump(merged, new=False)
            self._store[k] = merged
            self._emit("MODIFIED", merged)
            return copy.deepcopy(merged)

    def update(self, obj):
        obj = copy.deepcopy(obj)
        with self._lock:
            k = self._k(obj)
            cur = self._store.get(k)
            if cur is None:
                raise NotFound(str(k))
            if obj["metadata"].get("resourceVersion") not in (
                    None, cur["metadata"]["resourceVersion"]):
                raise Conflict(str(k))
            keep_status = cur.get("status")
            if keep_status is not None and "status" not in obj:
                obj["status"] = copy.deepcopy(keep_status)
            if obj.get("spec") != cur.get("spec"):
                obj.setdefault("metadata", {})["generation"] = \
                    int(cur["metadata"].get("generation", 1)) + 1
            else:
                obj["metadata"]["generation"] = \
                    cur["metadata"].get("generation", 1)
            obj["metadata"]["uid"] = cur["metadata"].get("uid")
            self._bump(obj, new=False)
            self._store[k] = obj
            self._emit("MODIFIED", obj)
            return copy.deepcopy(obj)

    def update_status(self, obj):
        with self._lock:
            k = self._k(obj)
            cur = self._store.get(k)
            if cur is None:
                raise NotFound(str(k))
            merged = copy.deepcopy(cur)
            merged["status"] = copy.deepcopy(obj.get("status") or {})
            self._bump(merged, new=False)
            self._store[k] = merged
            self._emit("MODIFIED", merged)
            return copy.deepcopy(merged)

    def patch(self, api_version, kind, namespace, name, patch):
        with self._lock:
            k = _key(api_version, kind, namespace, name)
            cur = self._store.get(k)
            if cur is None:
                raise NotFound(str(k))
            merged = copy.deepcopy(cur)
            _deep_merge(merged, patch)
            self._bump(merged, new=False)
            self._store[k] = merged
            self._emit("MODIFIED", merged)
            return copy.deepcopy(merged)

    def delete(self, api_version, kind, namespace, name):
        with self._lock:
            k = _key(api_version, kind, namespace, name)
            o = self._store.pop(k, None)
            if o is None:
                return False
            self._emit("DELETED", o)
            return True

    def watch(self, api_version, kind, namespace="", stop=None) -> Iterator[dict]:
        q: queue.Queue = queue.Queue()
        want = (api_version, kind, namespace)
        with self._lock:
            self._watchers.append((want, q))
            backlog = self.list(api_version, kind, namespace)
        for o in backlog:
            yield {"type": "ADDED", "object": o}
        while stop is None or not stop.is_set():
            try:
                yield q.get(timeout=0.05)
            except queue.Empty:
                if stop is None:
                    return
