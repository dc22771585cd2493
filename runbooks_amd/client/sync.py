"""Notebook file sync + port-forward.

Parity: reference internal/client/sync.go:28-135 (SyncFilesFromNotebook:
push the nbwatch agent into the pod, exec it, mirror WRITE/CREATE/REMOVE
events to the local dir) and port_forward.go:21-46.

The reference talks SPDY to the kubelet through client-go. Here the
NATIVE path (`client/remote.py`) speaks the same channel protocol over
the API server's WebSocket endpoints (v4.channel.k8s.io exec +
portforward) via aiohttp, used whenever a configured HTTPKubeClient is
passed; the kubectl subprocess path remains the kubeconfig-only
fallback (the reference itself shells out to kubectl for cp,
internal/cp/kubectl.go:15-26).
"""
from __future__ import annotations

import json
import os
import shutil
import subprocess
import threading
from typing import Callable, Optional

NBWATCH_POD_PATH = "/tmp/nbwatch.py"


def _kubectl() -> str:
    k = shutil.which("kubectl")
    if k is None:
        raise RuntimeError("kubectl not found on PATH (needed for sync/cp)")
    return k


def cp_to_pod(namespace: str, pod: str, src: str, dst: str,
              container: str = "notebook", client=None) -> None:
    if client is not None:
        from . import remote
        with open(src, "rb") as f:
            res = remote.cp_to_pod_native(client, namespace, pod, f.read(),
                                          dst, container=container)
        if res.returncode != 0:
            raise RuntimeError(f"cp_to_pod: {res.status}")
        return
    subprocess.run([_kubectl(), "cp", src, f"{namespace}/{pod}:{dst}",
                    "-c", container], check=True)


def cp_from_pod(namespace: str, pod: str, src: str, dst: str,
                container: str = "notebook", client=None) -> None:
    os.makedirs(os.path.dirname(dst) or ".", exist_ok=True)
    if client is not None:
        from . import remote
        data = remote.cp_from_pod_native(client, namespace, pod, src,
                                         container=container)
        with open(dst, "wb") as f:
            f.write(data)
        return
    subprocess.run([_kubectl(), "cp", f"{namespace}/{pod}:{src}", dst,
                    "-c", container], check=True)


def sync_files_from_notebook(namespace: str, pod: str, local_dir: str,
                             stop: Optional[threading.Event] = None,
                             on_event: Optional[Callable[[dict], None]] = None
                             ) -> None:
    """Copy the nbwatch agent into the pod, exec it, and mirror its events
    into local_dir until `stop` is set."""
    agent_src = os.path.join(os.path.dirname(os.path.dirname(__file__)),
                             "nbwatch.py")
    cp_to_pod(namespace, pod, agent_src, NBWATCH_POD_PATH)
    proc = subprocess.Popen(
        [_kubectl(), "exec", "-n", namespace, pod, "-c", "notebook", "--",
         "python3", NBWATCH_POD_PATH, "/content"],
        stdout=subprocess.PIPE, text=True)
    try:
        for line in proc.stdout:
            if stop is not None and stop.is_set():
                break
            line = line.strip()
            if not line:
                continue
            try:
                ev = json.loads(line)
            except json.JSONDecodeError:
                continue
            if on_event:
                on_event(ev)
            rel = os.path.relpath(ev["path"], "/content")
            local = os.path.join(local_dir, rel)
            if ev["op"] in ("WRITE", "CREATE", "RENAME"):
                cp_from_pod(namespace, pod, ev["path"], local)
            elif ev["op"] == "REMOVE":
                try:
                    os.remove(local)
                except FileNotFoundError:
                    pass
    finally:
        proc.terminate()


def port_forward(namespace: str, name: str, local_port: int, pod_port: int,
                 resource: str = "pod") -> subprocess.Popen:
    """Start a kubectl port-forward (pod or service); caller terminates
    the returned proc."""
    return subprocess.Popen(
        [_kubectl(), "port-forward", "-n", namespace, f"{resource}/{name}",
         f"{local_port}:{pod_port}"],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
