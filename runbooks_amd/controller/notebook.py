"""Notebook reconciler: Jupyter Pods with model/dataset mounts.

Parity: reference internal/controller/notebook_controller.go —
suspend → delete pod + Suspended condition (:134-155, :255-283 here),
model/dataset readiness gating (:160-250), Pod construction (:317-454:
default `jupyter lab --NotebookApp.token=$(NOTEBOOK_TOKEN)` command, port
8888, readiness GET /api, dataset/model RO + own artifacts RW mounts),
apply with delete-and-recreate on immutable-field conflicts (:266-281).
"""
from __future__ import annotations

from ..api import conditions as cond
from ..api.types import Dataset, Model, Notebook
from ..cloud import Mount, MountBucketConfig
from ..k8s import KubeClient
from .. import resources as res
from .params import ParamsReconciler, mount_params_config_map
from .utils import (
    Result,
    SA_NOTEBOOK,
    is_pod_ready,
    reconcile_service_account,
    resolve_env,
)


def nb_pod_name(nb: Notebook) -> str:
    return f"{nb.name}-notebook"


DEFAULT_COMMAND = [
    "jupyter", "lab", "--allow-root", "--ip=0.0.0.0",
    "--NotebookApp.token=$(NOTEBOOK_TOKEN)", "--notebook-dir=/content",
]


class NotebookReconciler:
    kind = "Notebook"

    def __init__(self, kube: KubeClient, cloud, sci_client):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci_client
        self.params = ParamsReconciler(kube)

    def reconcile(self, nb: Notebook) -> Result:
        if not nb.get_image():
            return Result()
        self.params.reconcile_params_config_map(nb)
        return self._reconcile_notebook(nb)

    def _gate(self, nb: Notebook, reason: str) -> Result:
        nb.set_status_ready(False)
        nb.set_condition(cond.CONDITION_SERVING, False, reason,
                         observed_generation=nb.generation)
        self.kube.update_status(nb.to_dict())
        return Result()

    def _reconcile_notebook(self, nb: Notebook) -> Result:
        if nb.is_suspended():
            nb.set_status_ready(False)
            nb.set_condition(cond.CONDITION_SERVING, False,
                             cond.REASON_SUSPENDED,
                             observed_generation=nb.generation)
            self.kube.update_status(nb.to_dict())
            self.kube.delete("v1", "Pod", nb.namespace, nb_pod_name(nb))
            return Result()

        reconcile_service_account(self.cloud, self.sci, self.kube,
                                  nb.namespace, SA_NOTEBOOK)

        model = None
        if nb.model is not None:
            raw = self.kube.get("substratus.ai/v1", "Model", nb.namespace,
                                nb.model.name)
            if raw is None:
                return self._gate(nb, cond.REASON_MODEL_NOT_FOUND)
            model = Model.from_dict(raw)
            if not model.ready:
                return self._gate(nb, cond.REASON_MODEL_NOT_READY)

        dataset = None
        if nb.dataset is not None:
            raw = self.kube.get("substratus.ai/v1", "Dataset", nb.namespace,
                                nb.dataset.name)
            if raw is None:
                return self._gate(nb, cond.REASON_DATASET_NOT_FOUND)
            dataset = Dataset.from_dict(raw)
            if not dataset.ready:
                return self._gate(nb, cond.REASON_DATASET_NOT_READY)

        pod = self._notebook_pod(nb, model, dataset)
        try:
            self.kube.apply(pod, field_manager="notebook-controller")
        except Exception:
            # Immutable-field conflict → delete & recreate; Pod event
            # requeues (reference notebook_controller.go:266-281).
            self.kube.delete("v1", "Pod", nb.namespace, nb_pod_name(nb))
            return Result()

        cur = self.kube.get("v1", "Pod", nb.namespace, nb_pod_name(nb)) or {}
        if is_pod_ready(cur):
            nb.set_status_ready(True)
            nb.set_condition(cond.CONDITION_SERVING, True,
                             cond.REASON_POD_READY,
                             observed_generation=nb.generation)
        else:
            nb.set_status_ready(False)
            nb.set_condition(cond.CONDITION_SERVING, False,
                             cond.REASON_POD_NOT_READY,
                             observed_generation=nb.generation)
        self.kube.update_status(nb.to_dict())
        return Result(success=True)

    def _notebook_pod(self, nb: Notebook, model, dataset) -> dict:
        container_name = "notebook"
        cmd = list(nb.command) or list(DEFAULT_COMMAND)
        env = resolve_env(nb.env)
        env.append({"name": "NOTEBOOK_TOKEN", "value": "default"})

        pod_meta = {
            "name": nb_pod_name(nb),
            "namespace": nb.namespace,
            "annotations": {
                "kubectl.kubernetes.io/default-container": container_name},
            "labels": {"notebook": nb.name, "role": "run"},
            "ownerReferences": [{
                "apiVersion": "substratus.ai/v1", "kind": "Notebook",
                "name": nb.name, "uid": nb.metadata.get("uid", ""),
                "controller": True, "blockOwnerDeletion": True}],
        }
        pod_spec = {
            "serviceAccountName": SA_NOTEBOOK,
            "containers": [{
                "name": container_name,
                "image": nb.get_image(),
                "command": cmd,
                "ports": [{"name": "notebook", "containerPort": 8888}],
                "env": env,
                "readinessProbe": {"httpGet": {"path": "/api",
                                               "port": 8888}},
            }],
        }
        mount_params_config_map(pod_spec, nb, container_name)
        if dataset is not None:
            self.cloud.mount_bucket(pod_meta, pod_spec, dataset,
                                    MountBucketConfig(
                                        name="dataset",
                                        container=container_name,
                                        mounts=[Mount("artifacts", "data")],
                                        read_only=True))
        if model is not None:
            self.cloud.mount_bucket(pod_meta, pod_spec, model,
                                    MountBucketConfig(
                                        name="model",
                                        container=container_name,
                                        mounts=[Mount("artifacts", "model")],
                                        read_only=True))
        self.cloud.mount_bucket(pod_meta, pod_spec, nb, MountBucketConfig(
            name="artifacts", container=container_name,
            mounts=[Mount("artifacts", "artifacts")], read_only=False))
        res.apply(pod_meta, pod_spec, container_name, self.cloud.name,
                  nb.resources)
        return {"apiVersion": "v1", "kind": "Pod", "metadata": pod_meta,
                "spec": pod_spec}
