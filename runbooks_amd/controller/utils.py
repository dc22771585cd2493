"""Shared reconcile helpers (parity: reference internal/controller/utils.go).
"""
from __future__ import annotations

import re
from dataclasses import dataclass

from ..k8s import Conflict, KubeClient


@dataclass
class Result:
    """Reconcile outcome propagated up the call stack
    (reference utils.go:17-21)."""
    success: bool = False
    failure: bool = False
    requeue_after: float = 0.0


def job_result(job: dict) -> tuple[bool, bool]:
    """(complete, failed) from Job status conditions
    (reference utils.go:36-49)."""
    for c in (job.get("status") or {}).get("conditions") or []:
        if c.get("type") == "Complete" and c.get("status") == "True":
            return True, False
        if c.get("type") == "Failed" and c.get("status") == "True":
            return False, True
    return False, False


def reconcile_job(kube: KubeClient, job: dict) -> Result:
    """Create-if-absent, then read completion (reference utils.go:23-34)."""
    try:
        kube.create(job)
    except Conflict:
        pass
    m = job["metadata"]
    cur = kube.get(job["apiVersion"], "Job", m.get("namespace", "default"),
                   m["name"])
    if cur is None:
        return Result()
    complete, failed = job_result(cur)
    return Result(success=complete, failure=failed)


def is_pod_ready(pod: dict) -> bool:
    """(reference utils.go:51-65)"""
    status = pod.get("status") or {}
    if status.get("phase") != "Running":
        return False
    for c in status.get("conditions") or []:
        if c.get("type") == "Ready" and c.get("status") == "True":
            return True
    return False


_SECRET_RE = re.compile(r"\${{ *secrets\.(.+)\.(.+?) *}}")


def resolve_env(env: dict[str, str]) -> list[dict]:
    """Map spec.env to container env vars, expanding the GitHub-actions-style
    `${{ secrets.name.key }}` syntax to SecretKeyRef
    (reference utils.go:67-93)."""
    out = []
    for key in sorted(env):
        value = env[key]
        m = _SECRET_RE.search(value)
        if m:
            name, k = m.group(1).strip(), m.group(2).strip()
            out.append({"name": key, "valueFrom": {"secretKeyRef": {
                "name": name, "key": k}}})
        else:
            out.append({"name": key, "value": value})
    return out


def container_by_name(pod_spec: dict, name: str) -> dict:
    for group in ("containers", "initContainers"):
        for c in pod_spec.get(group, []):
            if c["name"] == name:
                return c
    raise ValueError(f"container not found: {name}")


# role service-account names
# (reference service_accounts_controller.go:16-22)
SA_CONTAINER_BUILDER = "container-builder"
SA_MODELLER = "modeller"
SA_MODEL_SERVER = "model-server"
SA_NOTEBOOK = "notebook"
SA_DATA_LOADER = "data-loader"


def reconcile_service_account(cloud, sci_client, kube: KubeClient,
                              namespace: str, name: str) -> Result:
    """Ensure the role SA exists and its cloud principal is bound
    (reference service_accounts_controller.go:38-66)."""
    sa = kube.get("v1", "ServiceAccount", namespace, name) or {
        "apiVersion": "v1", "kind": "ServiceAccount",
        "metadata": {"name": name, "namespace": namespace}}
    principal, bound = cloud.get_principal(sa)
    if not bound:
        sci_client.bind_identity(kubernetes_service_account=name,
                                 kubernetes_namespace=namespace,
                                 principal=principal)
    cloud.associate_principal(sa)
    kube.apply(sa)
    return Result(success=True)
