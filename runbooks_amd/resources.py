"""Compute-resource application and the AMD GPU scheduling table.

Parity: reference internal/resources/resources.go:13-72 (Apply),
:74-91 (ContainerBuilderResources) and gpu_info.go:15-48 (GPU table) — with
the NVIDIA resource name / GKE accelerator selectors replaced by the ROCm
k8s-device-plugin's `amd.com/gpu` resource and an AMD product node label.
NVIDIA GPUType values from reference manifests are accepted and mapped onto
the MI355X pool so the reference's example YAMLs schedule unchanged.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

from .api.types import Resources

GIGABYTE = 1024 ** 3

AMD_GPU_RESOURCE = "amd.com/gpu"
# Label published by the ROCm k8s-device-plugin's node labeller.
AMD_PRODUCT_LABEL = "amd.com/gpu.product"


@dataclass
class GPUInfo:
    resource_name: str
    node_selector: dict[str, str] = field(default_factory=dict)


_MI355X = GPUInfo(AMD_GPU_RESOURCE, {AMD_PRODUCT_LABEL: "MI355X"})
_MI300X = GPUInfo(AMD_GPU_RESOURCE, {AMD_PRODUCT_LABEL: "MI300X"})

_CLOUD_GPUS: dict[str, dict[str, GPUInfo]] = {
    "gcp": {
        "amd-mi355x": _MI355X,
        "amd-mi300x": _MI300X,
        # reference manifests name NVIDIA parts (common_types.go:94-100);
        # on this platform they all land on the MI355X pool.
        "nvidia-t4": _MI355X,
        "nvidia-l4": _MI355X,
        "nvidia-a100": _MI355X,
    },
}


def get_gpu_info(cloud_name: str, gpu_type: str) -> Optional[GPUInfo]:
    """(reference gpu_info.go:15-23; kind passes through with no selector)"""
    if cloud_name == "kind":
        return GPUInfo(AMD_GPU_RESOURCE, {})
    return _CLOUD_GPUS.get(cloud_name, {}).get(gpu_type)


def apply(pod_metadata: dict, pod_spec: dict, container_name: str,
          cloud_name: str, res: Optional[Resources]) -> None:
    """Set requests/limits + GPU resource + node selector on the named
    container (reference resources.go:13-72)."""
    if res is None:
        res = Resources(cpu=0, disk=0, memory=0) if cloud_name == "kind" \
            else Resources(cpu=2, memory=4, disk=100)

    requests: dict[str, str] = {}
    limits: dict[str, str] = {}
    if res.cpu:
        requests["cpu"] = str(res.cpu)
    if res.memory:
        requests["memory"] = f"{res.memory}Gi"
    if res.disk:
        requests["ephemeral-storage"] = f"{res.disk}Gi"

    if res.gpu is not None and res.gpu.count:
        info = get_gpu_info(cloud_name, res.gpu.type or "amd-mi355x")
        if info is None:
            raise ValueError(
                f"GPU {res.gpu.type} is not supported on cloud {cloud_name}")
        requests[info.resource_name] = str(res.gpu.count)
        limits[info.resource_name] = str(res.gpu.count)
        sel = pod_spec.setdefault("nodeSelector", {})
        sel.update(info.node_selector)
        # Spot toleration triggers node auto-provisioning on GKE
        # (reference resources.go:56-62); harmless elsewhere.
        pod_spec.setdefault("tolerations", []).append({
            "key": "cloud.google.com/gke-spot", "operator": "Equal",
            "value": "true", "effect": "NoSchedule"})
        # RCCL over xGMI needs IPC between the per-GPU ranks in the pod.
        pod_spec.setdefault("shareProcessNamespace", True)

    if not _set_container_resources(container_name, pod_spec, requests,
                                    limits):
        raise ValueError(f"container {container_name} not found in pod")


def container_builder_resources(cloud_name: str) -> dict:
    """(reference resources.go:74-91)"""
    if cloud_name == "kind":
        return {}
    return {"requests": {"cpu": "2", "memory": "12Gi",
                         "ephemeral-storage": "100Gi"}}


def _set_container_resources(name: str, pod_spec: dict, requests: dict,
                             limits: dict) -> bool:
    for group in ("initContainers", "containers"):
        for c in pod_spec.get(group, []):
            if c["name"] == name:
                r = c.setdefault("resources", {})
                r.setdefault("requests", {}).update(requests)
                r.setdefault("limits", {}).update(limits)
                return True
    return False
