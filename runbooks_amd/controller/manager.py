"""Controller manager: wires the reconcilers to watch events.

Parity: reference cmd/controllermanager/main.go:40-241 (flags, cloud
autodetect, SCI dial, registration of the four kind controllers + four
BuildReconcilers + SA principal association, healthz) and
internal/controller/manager.go:14-72 (field indexes on spec.model.name /
spec.dataset.name that map dependency events to dependent objects).
"""
from __future__ import annotations

import argparse
import logging
import queue
import threading
from typing import Optional

from ..api.types import KINDS
from ..k8s import HTTPKubeClient, KubeClient
from .build import BuildReconciler
from .dataset import DatasetReconciler
from .model import ModelReconciler
from .notebook import NotebookReconciler
from .server import ServerReconciler
from .utils import reconcile_service_account

log = logging.getLogger("runbooks_amd.controller")

API = "substratus.ai/v1"


try:
    from prometheus_client import Counter, Histogram
    _M_RECONCILES = Counter("rb_reconciles_total", "reconcile calls",
                            ["kind"])
    _M_ERRORS = Counter("rb_reconcile_errors_total", "reconcile errors",
                        ["kind"])
    _M_LAT = Histogram("rb_reconcile_seconds", "reconcile latency",
                       ["kind"],
                       buckets=(.001, .005, .01, .05, .1, .5, 1, 5))
except ImportError:  # pragma: no cover
    _M_RECONCILES = _M_ERRORS = _M_LAT = None


class ControllerManager:
    def __init__(self, kube: KubeClient, cloud, sci_client):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci_client
        self.reconcilers = {
            "Model": ModelReconciler(kube, cloud, sci_client),
            "Dataset": DatasetReconciler(kube, cloud, sci_client),
            "Server": ServerReconciler(kube, cloud, sci_client),
            "Notebook": NotebookReconciler(kube, cloud, sci_client),
        }
        # One BuildReconciler per kind (reference main.go:142-224).
        self.builders = {
            kind: BuildReconciler(kube, cloud, sci_client, kind, cls)
            for kind, cls in KINDS.items()
        }
        self._queue: "queue.Queue[tuple[str, str, str]]" = queue.Queue()
        self._stop = threading.Event()

    # -- single-object reconcile -------------------------------------------
    def reconcile_object(self, kind: str, namespace: str, name: str) -> None:
        import time as _time
        t0 = _time.perf_counter()
        try:
            self._reconcile_object(kind, namespace, name)
        except Exception:
            if _M_ERRORS is not None:
                _M_ERRORS.labels(kind).inc()
            raise
        finally:
            if _M_RECONCILES is not None:
                _M_RECONCILES.labels(kind).inc()
                _M_LAT.labels(kind).observe(_time.perf_counter() - t0)

    def _reconcile_object(self, kind: str, namespace: str, name: str) -> None:
        raw = self.kube.get(API, kind, namespace, name)
        if raw is None:
            return
        obj = KINDS[kind].from_dict(raw)
        if obj.get_build() is not None and \
                obj.get_image() != self.cloud.object_built_image_url(obj):
            self.builders[kind].reconcile(obj)
            raw = self.kube.get(API, kind, namespace, name)
            if raw is None:
                return
            obj = KINDS[kind].from_dict(raw)
        self.reconcilers[kind].reconcile(obj)

    # -- dependency fan-out (field indexes, reference manager.go:23-72) ----
    def _dependents(self, kind: str, namespace: str,
                    name: str) -> list[tuple[str, str, str]]:
        out = []
        if kind == "Model":
            for m in self.kube.list(API, "Model", namespace):
                if ((m["spec"].get("model") or {}).get("name")) == name:
                    out.append(("Model", namespace, m["metadata"]["name"]))
            for s in self.kube.list(API, "Server", namespace):
                if ((s["spec"].get("model") or {}).get("name")) == name:
                    out.append(("Server", namespace, s["metadata"]["name"]))
            for n in self.kube.list(API, "Notebook", namespace):
                if ((n["spec"].get("model") or {}).get("name")) == name:
                    out.append(("Notebook", namespace, n["metadata"]["name"]))
        elif kind == "Dataset":
            for m in self.kube.list(API, "Model", namespace):
                if ((m["spec"].get("dataset") or {}).get("name")) == name:
                    out.append(("Model", namespace, m["metadata"]["name"]))
            for n in self.kube.list(API, "Notebook", namespace):
                if ((n["spec"].get("dataset") or {}).get("name")) == name:
                    out.append(("Notebook", namespace, n["metadata"]["name"]))
        return out

    def _owner_request(self, obj: dict) -> Optional[tuple[str, str, str]]:
        for ref in obj["metadata"].get("ownerReferences", []):
            if ref.get("apiVersion") == API and ref.get("kind") in KINDS:
                return (ref["kind"],
                        obj["metadata"].get("namespace", "default"),
                        ref["name"])
        return None

    # -- test-friendly settle loop (plays the role of envtest's
    #    Eventually(...) + the watch-driven workqueue) ----------------------
    def reconcile_all(self, namespace: str = "", rounds: int = 6) -> None:
        for _ in range(rounds):
            for kind in KINDS:
                for raw in self.kube.list(API, kind, namespace):
                    m = raw["metadata"]
                    try:
                        self.reconcile_object(kind, m["namespace"], m["name"])
                    except Exception:
                        log.exception("reconcile %s/%s failed", kind,
                                      m["name"])

    # -- production loop ----------------------------------------------------
    def run(self) -> None:
        threads = []
        for kind in KINDS:
            threads.append(threading.Thread(
                target=self._watch_kind, args=(API, kind, False),
                daemon=True))
        for api, kind in (("batch/v1", "Job"), ("v1", "Pod"),
                          ("apps/v1", "Deployment")):
            threads.append(threading.Thread(
                target=self._watch_kind, args=(api, kind, True),
                daemon=True))
        for t in threads:
            t.start()
        while not self._stop.is_set():
            try:
                kind, ns, name = self._queue.get(timeout=0.5)
            except queue.Empty:
                continue
            try:
                self.reconcile_object(kind, ns, name)
                for req in self._dependents(kind, ns, name):
                    self._queue.put(req)
            except Exception:
                log.exception("reconcile %s/%s/%s failed", kind, ns, name)

    def stop(self) -> None:
        self._stop.set()

    def _watch_kind(self, api: str, kind: str, owned: bool) -> None:
        while not self._stop.is_set():
            try:
                # Level-triggered re-list on every watch (re)establishment:
                # anything created/updated while the watch was down would
                # otherwise never reconcile (controller-runtime's informer
                # does the same list+watch dance; reference
                # internal/controller relies on it implicitly).
                for obj in self.kube.list(api, kind):
                    self._enqueue(kind, obj, owned)
                for ev in self.kube.watch(api, kind, stop=self._stop):
                    self._enqueue(kind, ev["object"], owned)
            except Exception:
                log.exception("watch %s restarting", kind)
                self._stop.wait(1.0)

    def _enqueue(self, kind: str, obj: dict, owned: bool) -> None:
        if owned:
            req = self._owner_request(obj)
            if req:
                self._queue.put(req)
        else:
            m = obj["metadata"]
            self._queue.put((kind, m.get("namespace", "default"), m["name"]))


class LeaderElector:
    """Lease-based leader election (parity: the reference manager's
    --leader-elect flag, reference cmd/controllermanager/main.go:67-68,
    backed by controller-runtime's coordination.k8s.io Lease lock)."""

    def __init__(self, kube: KubeClient, identity: str,
                 namespace: str = "substratus",
                 name: str = "runbooks-amd-controller-manager",
                 lease_seconds: int = 15):
        self.kube = kube
        self.identity = identity
        self.namespace = namespace
        self.name = name
        self.lease_seconds = lease_seconds

    def _now(self) -> str:
        import datetime
        return datetime.datetime.now(datetime.timezone.utc).strftime(
            "%Y-%m-%dT%H:%M:%S.%f")[:-3] + "Z"

    def _lease_obj(self) -> dict:
        return {
            "apiVersion": "coordination.k8s.io/v1", "kind": "Lease",
            "metadata": {"name": self.name, "namespace": self.namespace},
            "spec": {"holderIdentity": self.identity,
                     "leaseDurationSeconds": self.lease_seconds,
                     "renewTime": self._now()},
        }

    def try_acquire(self) -> bool:
        import datetime
        cur = self.kube.get("coordination.k8s.io/v1", "Lease",
                            self.namespace, self.name)
        if cur is None:
            try:
                self.kube.create(self._lease_obj())
                return True
            except Exception:
                return False
        spec = cur.get("spec") or {}
        holder = spec.get("holderIdentity")
        if holder == self.identity:
            self.renew()
            return True
        renew = spec.get("renewTime", "")
        try:
            t = datetime.datetime.strptime(renew[:19], "%Y-%m-%dT%H:%M:%S")
            age = (datetime.datetime.utcnow() - t).total_seconds()
        except ValueError:
            age = 1e9
        if age > spec.get("leaseDurationSeconds", self.lease_seconds) * 2:
            self.kube.patch("coordination.k8s.io/v1", "Lease",
                            self.namespace, self.name,
                            {"spec": self._lease_obj()["spec"]})
            return True
        return False

    def renew(self) -> None:
        self.kube.patch("coordination.k8s.io/v1", "Lease", self.namespace,
                        self.name, {"spec": {"renewTime": self._now()}})

    def run(self, stop: threading.Event) -> None:
        """Block until leadership is acquired; keep renewing in the
        background."""
        while not stop.is_set() and not self.try_acquire():
            stop.wait(2.0)

        def _renew_loop():
            while not stop.is_set():
                stop.wait(self.lease_seconds / 3)
                try:
                    self.renew()
                except Exception:
                    log.exception("lease renew failed")

        threading.Thread(target=_renew_loop, daemon=True).start()


def run_manager(argv: Optional[list[str]] = None) -> None:
    """controllermanager entrypoint (reference cmd/controllermanager/main.go).
    """
    from ..cloud import new_cloud
    from ..sci import ControllerClient

    p = argparse.ArgumentParser(description="runbooks-amd controller manager")
    p.add_argument("--sci-address", default="sci.substratus.svc.cluster.local:10080")
    p.add_argument("--health-probe-bind-address", default=":8081")
    p.add_argument("--namespace", default="")
    p.add_argument("--leader-elect", action="store_true")
    args = p.parse_args(argv)

    logging.basicConfig(level=logging.INFO)
    kube = HTTPKubeClient()
    cloud = new_cloud()
    sci_client = ControllerClient(args.sci_address)

    if args.leader_elect:
        import os as _os
        import socket
        stop = threading.Event()
        LeaderElector(kube, f"{socket.gethostname()}_{_os.getpid()}").run(stop)

    # Associate the SCI server's own service account principal
    # (reference main.go:117-127).
    try:
        reconcile_service_account(cloud, sci_client, kube, "substratus", "sci")
    except Exception:
        log.exception("associating SCI service account principal")

    mgr = ControllerManager(kube, cloud, sci_client)
    _serve_health(args.health_probe_bind_address)
    mgr.run()


def _serve_health(addr: str) -> None:
    """healthz/readyz endpoints (reference main.go:227-234)."""
    import http.server

    host, _, port = addr.rpartition(":")

    class H(http.server.BaseHTTPRequestHandler):
        def do_GET(self):
            if self.path == "/metrics":
                try:
                    from prometheus_client import (
                        CONTENT_TYPE_LATEST,
                        generate_latest,
                    )
                    body = generate_latest()
                    self.send_response(200)
                    self.send_header("Content-Type", CONTENT_TYPE_LATEST)
                    self.send_header("Content-Length", str(len(body)))
                    self.end_headers()
                    self.wfile.write(body)
                    return
                except ImportError:
                    pass
            code = 200 if self.path in ("/healthz", "/readyz") else 404
            self.send_response(code)
            self.end_headers()

        def log_message(self, *a):
            pass

    httpd = http.server.ThreadingHTTPServer((host or "0.0.0.0", int(port)), H)
    threading.Thread(target=httpd.serve_forever, daemon=True).start()


if __name__ == "__main__":
    run_manager()
