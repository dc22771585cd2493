"""Dataset reconciler: containerized data-loader Jobs.

Parity: reference internal/controller/dataset_controller.go —
Reconcile (:35-60), loadJob (:149-217): `{name}-data-loader` Job, the
`data-loader` SA, artifacts RW mount, backoffLimit=2.
"""
from __future__ import annotations

from ..api import conditions as cond
from ..api.types import Dataset
from ..cloud import Mount, MountBucketConfig
from ..k8s import KubeClient
from .. import resources as res
from .params import ParamsReconciler, mount_params_config_map
from .utils import (
    Result,
    SA_DATA_LOADER,
    reconcile_job,
    reconcile_service_account,
    resolve_env,
)


class DatasetReconciler:
    kind = "Dataset"

    def __init__(self, kube: KubeClient, cloud, sci_client):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci_client
        self.params = ParamsReconciler(kube)

    def reconcile(self, ds: Dataset) -> Result:
        if not ds.get_image():
            return Result()
        self.params.reconcile_params_config_map(ds)

        if ds.ready:
            return Result(success=True)

        ds.artifacts.url = str(self.cloud.object_artifact_url(ds))
        reconcile_service_account(self.cloud, self.sci, self.kube,
                                  ds.namespace, SA_DATA_LOADER)

        job = self._load_job(ds)
        jr = reconcile_job(self.kube, job)
        if not jr.success:
            ds.set_status_ready(False)
            reason = cond.REASON_JOB_FAILED if jr.failure else \
                cond.REASON_JOB_NOT_COMPLETE
            ds.set_condition(cond.CONDITION_COMPLETE, False, reason,
                             "Waiting for data-loader Job to complete"
                             if not jr.failure else "",
                             ds.generation)
            self.kube.update_status(ds.to_dict())
            return jr

        ds.set_status_ready(True)
        ds.set_condition(cond.CONDITION_COMPLETE, True,
                         cond.REASON_JOB_COMPLETE,
                         observed_generation=ds.generation)
        self.kube.update_status(ds.to_dict())
        return Result(success=True)

    def _load_job(self, ds: Dataset) -> dict:
        container_name = "loader"
        pod_meta = {
            "annotations": {
                "kubectl.kubernetes.io/default-container": container_name},
            "labels": {"dataset": ds.name, "role": "run"},
        }
        pod_spec = {
            "securityContext": {"fsGroup": 3003},
            "serviceAccountName": SA_DATA_LOADER,
            "containers": [{
                "name": container_name,
                "image": ds.get_image(),
                "command": list(ds.command),
                "env": resolve_env(ds.env),
            }],
            "restartPolicy": "Never",
        }
        mount_params_config_map(pod_spec, ds, container_name)
        self.cloud.mount_bucket(pod_meta, pod_spec, ds, MountBucketConfig(
            name="artifacts", container=container_name,
            mounts=[Mount("artifacts", "artifacts")], read_only=False))
        res.apply(pod_meta, pod_spec, container_name, self.cloud.name,
                  ds.resources)
        return {
            "apiVersion": "batch/v1", "kind": "Job",
            "metadata": {
                "name": f"{ds.name}-data-loader",
                "namespace": ds.namespace,
                "ownerReferences": [{
                    "apiVersion": "substratus.ai/v1", "kind": "Dataset",
                    "name": ds.name, "uid": ds.metadata.get("uid", ""),
                    "controller": True, "blockOwnerDeletion": True}],
            },
            "spec": {
                "backoffLimit": 2,
                "template": {"metadata": pod_meta, "spec": pod_spec},
            },
        }
