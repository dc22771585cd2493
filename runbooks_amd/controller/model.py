"""Model reconciler: import / fine-tune Jobs.

Parity: reference internal/controller/model_controller.go —
Reconcile (:43-68), readiness gating on base Model and Dataset (:96-172),
modellerJob construction (:286-395) with artifact RW + dataset/model RO
bucket mounts, params ConfigMap mount, AMD GPU resources, and the
backoffLimit heuristic (:294-303: cheap CPU import jobs retry, GPU jobs
don't).
"""
from __future__ import annotations

from ..api import conditions as cond
from ..api.types import Dataset, Model
from ..cloud import Mount, MountBucketConfig
from ..k8s import KubeClient
from .. import resources as res
from .params import ParamsReconciler, mount_params_config_map
from .utils import (
    Result,
    SA_MODELLER,
    reconcile_job,
    reconcile_service_account,
    resolve_env,
)


class ModelReconciler:
    kind = "Model"

    def __init__(self, kube: KubeClient, cloud, sci_client):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci_client
        self.params = ParamsReconciler(kube)

    def reconcile(self, model: Model) -> Result:
        # The BuildReconciler owns objects with no image yet
        # (reference model_controller.go:54-57).
        if not model.get_image():
            return Result()
        self.params.reconcile_params_config_map(model)
        return self._reconcile_model(model)

    def _gate(self, model: Model, reason: str) -> Result:
        model.set_status_ready(False)
        model.set_condition(cond.CONDITION_COMPLETE, False, reason,
                            observed_generation=model.generation)
        self.kube.update_status(model.to_dict())
        return Result()  # watch-driven requeue

    def _reconcile_model(self, model: Model) -> Result:
        if model.ready:
            return Result(success=True)

        model.artifacts.url = str(self.cloud.object_artifact_url(model))

        reconcile_service_account(self.cloud, self.sci, self.kube,
                                  model.namespace, SA_MODELLER)

        base_model = None
        if model.model is not None:
            raw = self.kube.get("substratus.ai/v1", "Model", model.namespace,
                                model.model.name)
            if raw is None:
                return self._gate(model, cond.REASON_BASE_MODEL_NOT_FOUND)
            base_model = Model.from_dict(raw)
            if not base_model.ready:
                return self._gate(model, cond.REASON_BASE_MODEL_NOT_READY)

        dataset = None
        if model.dataset is not None:
            raw = self.kube.get("substratus.ai/v1", "Dataset",
                                model.namespace, model.dataset.name)
            if raw is None:
                return self._gate(model, cond.REASON_DATASET_NOT_FOUND)
            dataset = Dataset.from_dict(raw)
            if not dataset.ready:
                return self._gate(model, cond.REASON_DATASET_NOT_READY)

        job = self._modeller_job(model, base_model, dataset)
        jr = reconcile_job(self.kube, job)
        if not jr.success:
            model.set_status_ready(False)
            if jr.failure:
                model.set_condition(cond.CONDITION_COMPLETE, False,
                                    cond.REASON_JOB_FAILED,
                                    observed_generation=model.generation)
            else:
                model.set_condition(cond.CONDITION_COMPLETE, False,
                                    cond.REASON_JOB_NOT_COMPLETE,
                                    "Waiting for modeller Job to complete",
                                    model.generation)
            self.kube.update_status(model.to_dict())
            return jr

        model.set_status_ready(True)
        model.set_condition(cond.CONDITION_COMPLETE, True,
                            cond.REASON_JOB_COMPLETE,
                            observed_generation=model.generation)
        self.kube.update_status(model.to_dict())
        return Result(success=True)

    def _modeller_job(self, model: Model, base_model, dataset) -> dict:
        container_name = "model"
        # Expensive (GPU) jobs don't retry; cheap CPU import jobs do
        # (reference model_controller.go:294-303).
        backoff = 0
        if (model.resources is not None and model.resources.cpu <= 3 and
                model.resources.gpu is not None and
                model.resources.gpu.count == 0):
            backoff = 2

        pod_meta = {
            "annotations": {
                "kubectl.kubernetes.io/default-container": container_name},
            "labels": {"model": model.name, "role": "run"},
        }
        pod_spec = {
            "securityContext": {"fsGroup": 3003},
            "serviceAccountName": SA_MODELLER,
            "containers": [{
                "name": container_name,
                "image": model.get_image(),
                "command": list(model.command),
                "env": resolve_env(model.env),
            }],
            "restartPolicy": "Never",
        }
        mount_params_config_map(pod_spec, model, container_name)
        self.cloud.mount_bucket(pod_meta, pod_spec, model, MountBucketConfig(
            name="artifacts", container=container_name,
            mounts=[Mount("artifacts", "artifacts")], read_only=False))
        if dataset is not None:
            self.cloud.mount_bucket(pod_meta, pod_spec, dataset,
                                    MountBucketConfig(
                                        name="dataset",
                                        container=container_name,
                                        mounts=[Mount("artifacts", "data")],
                                        read_only=True))
        if base_model is not None:
            self.cloud.mount_bucket(pod_meta, pod_spec, base_model,
                                    MountBucketConfig(
                                        name="model",
                                        container=container_name,
                                        mounts=[Mount("artifacts", "model")],
                                        read_only=True))
        res.apply(pod_meta, pod_spec, container_name, self.cloud.name,
                  model.resources)
        return {
            "apiVersion": "batch/v1", "kind": "Job",
            "metadata": {
                "name": f"{model.name}-modeller",
                "namespace": model.namespace,
                "ownerReferences": [{
                    "apiVersion": "substratus.ai/v1", "kind": "Model",
                    "name": model.name,
                    "uid": model.metadata.get("uid", ""),
                    "controller": True, "blockOwnerDeletion": True}],
            },
            "spec": {
                "backoffLimit": backoff,
                "template": {"metadata": pod_meta, "spec": pod_spec},
            },
        }
