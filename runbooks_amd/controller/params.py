"""Params reconciler: .spec.params → ConfigMap + /content/params.json mount.

Parity: reference internal/controller/params_reconciler.go:23-104. The
documented PARAM_* env conversion happens in the workload images' entrypoint
(images/entrypoint.py — the reference leaves it to its external images,
reference container-contract.md:34-48).
"""
from __future__ import annotations

import json

from ..k8s import KubeClient
from .utils import Result


def params_config_map_name(obj) -> str:
    """{name}-{kind}-params (reference params_reconciler.go:70-76)."""
    if not obj.kind:
        raise ValueError("empty kind")
    return f"{obj.name}-{obj.kind.lower()}-params"


class ParamsReconciler:
    def __init__(self, kube: KubeClient):
        self.kube = kube

    def reconcile_params_config_map(self, obj) -> Result:
        params = obj.get_params()
        contents = json.dumps(params, indent=2) if params else "{}"
        cm = {
            "apiVersion": "v1", "kind": "ConfigMap",
            "metadata": {"name": params_config_map_name(obj),
                         "namespace": obj.namespace,
                         "labels": {"app.kubernetes.io/managed-by":
                                    "runbooks-amd"}},
            "data": {"params.json": contents},
        }
        self.kube.apply(cm)
        return Result(success=True)


def mount_params_config_map(pod_spec: dict, obj, container: str) -> None:
    """SubPath-mount params.json at /content/params.json
    (reference params_reconciler.go:78-104)."""
    pod_spec.setdefault("volumes", []).append({
        "name": "params",
        "configMap": {"name": params_config_map_name(obj)},
    })
    for c in pod_spec.get("containers", []):
        if c["name"] == container:
            c.setdefault("volumeMounts", []).append({
                "name": "params",
                "mountPath": "/content/params.json",
                "subPath": "params.json",
            })
            return
    raise ValueError(f"container not found: {container}")
