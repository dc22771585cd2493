"""Generic image-build reconciler, instantiated once per kind.

Parity: reference internal/controller/build_reconciler.go:59-174 (state
machine), :183-268 (signed-URL upload handshake), :270-403 (git → kaniko
Job), :405-533 (storage-context kaniko Job with the kind tar:// hostPath
hack), :535-580 (SCI calls + job naming).
"""
from __future__ import annotations

import datetime
from typing import Callable

from ..api.types import UploadStatus
from ..api import conditions as cond
from ..k8s import Conflict, KubeClient
from ..resources import container_builder_resources
from .utils import (
    Result,
    SA_CONTAINER_BUILDER,
    reconcile_service_account,
)

LATEST_UPLOAD_PATH = "uploads/latest.tar.gz"
BUILDER_IMAGE = "gcr.io/kaniko-project/executor:latest"
GIT_IMAGE = "alpine/git"


def build_job_name(obj) -> str:
    """{name}-{kind}-bld (reference build_reconciler.go:576-580)."""
    return f"{obj.name}-{obj.kind.lower()}-bld"


class BuildReconciler:
    def __init__(self, kube: KubeClient, cloud, sci_client,
                 kind: str, new_object: Callable):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci_client
        self.kind = kind
        self.new_object = new_object

    def reconcile(self, obj) -> Result:
        build = obj.get_build()
        if build is None:
            return Result(success=True)
        image = self.cloud.object_built_image_url(obj)
        if obj.get_image() == image:
            return Result(success=True)

        reconcile_service_account(self.cloud, self.sci, self.kube,
                                  obj.namespace, SA_CONTAINER_BUILDER)

        if build.upload is not None:
            r = self._reconcile_upload_file(obj)
            if not r.success:
                return r
            job = self._storage_build_job(obj)
        elif build.git is not None:
            job = self._git_build_job(obj)
        else:
            return Result()

        cur = self.kube.get("batch/v1", "Job", obj.namespace,
                            job["metadata"]["name"])
        if cur is None:
            try:
                self.kube.create(job)
            except Conflict:
                pass
            cur = self.kube.get("batch/v1", "Job", obj.namespace,
                                job["metadata"]["name"])
        if cur and cur["metadata"].get("annotations", {}).get("image") != image:
            # Out of date, recreate (reference build_reconciler.go:128-136).
            self.kube.delete("batch/v1", "Job", obj.namespace,
                             job["metadata"]["name"])
            self.kube.create(job)
            cur = job

        succeeded = int(((cur or {}).get("status") or {})
                        .get("succeeded", 0) or 0)
        if succeeded < 1:
            obj.set_status_ready(False)
            obj.set_condition(cond.CONDITION_BUILT, False,
                              cond.REASON_JOB_NOT_COMPLETE,
                              f"Waiting for builder Job to complete: "
                              f"{job['metadata']['name']}",
                              obj.generation)
            self.kube.update_status(obj.to_dict())
            return Result()  # Job watch requeues

        obj.set_image(image)
        self.kube.patch("substratus.ai/v1", obj.kind, obj.namespace,
                        obj.name, {"spec": {"image": image}})
        obj.set_condition(cond.CONDITION_BUILT, True,
                          cond.REASON_JOB_COMPLETE,
                          f"Builder Job completed: {job['metadata']['name']}",
                          obj.generation)
        self.kube.update_status(obj.to_dict())
        return Result(success=True)

    # -- upload handshake ---------------------------------------------------
    def _reconcile_upload_file(self, obj) -> Result:
        spec = obj.get_build().upload
        status = obj.get_status_upload()

        if spec.request_id != status.request_id:
            # Edge case: a matching upload may already exist in storage
            # (reference build_reconciler.go:192-210).
            existing = self._storage_object_md5(obj)
            if existing and existing == spec.md5_checksum:
                obj.set_status_upload(UploadStatus(
                    stored_md5_checksum=spec.md5_checksum))
                obj.set_condition(
                    cond.CONDITION_UPLOADED, True, cond.REASON_UPLOAD_FOUND,
                    f"Existing upload found in storage with specified "
                    f"checksum: {spec.md5_checksum}", obj.generation)
                self.kube.update_status(obj.to_dict())
                return Result(success=True)

            url, expiration = self._generate_signed_url(obj)
            obj.set_status_upload(UploadStatus(
                signed_url=url, request_id=spec.request_id,
                expiration=expiration))
            obj.set_condition(
                cond.CONDITION_UPLOADED, False, cond.REASON_AWAITING_UPLOAD,
                f"Waiting for upload with md5 checksum: {spec.md5_checksum}",
                obj.generation)
            self.kube.update_status(obj.to_dict())
            # Client triggers a change after uploading → requeue.
            return Result()

        storage_md5 = self._storage_object_md5(obj)
        if storage_md5 != spec.md5_checksum:
            # Upload may be in progress; client retriggers via annotation.
            return Result()

        obj.set_status_upload(UploadStatus(
            request_id=spec.request_id, stored_md5_checksum=storage_md5))
        obj.set_condition(
            cond.CONDITION_UPLOADED, True, cond.REASON_UPLOAD_FOUND,
            f"Upload received with matching md5 checksum: "
            f"{spec.md5_checksum}", obj.generation)
        self.kube.update_status(obj.to_dict())
        return Result(success=True)

    def _storage_object_md5(self, obj) -> str:
        u = self.cloud.object_artifact_url(obj)
        try:
            resp = self.sci.get_object_md5(
                bucket_name=u.bucket,
                object_name=f"{u.path}/{LATEST_UPLOAD_PATH}")
            return resp.md5_checksum
        except Exception:
            return ""

    def _generate_signed_url(self, obj) -> tuple[str, str]:
        u = self.cloud.object_artifact_url(obj)
        expiration_seconds = 300
        resp = self.sci.create_signed_url(
            bucket_name=u.bucket,
            object_name=f"{u.path}/{LATEST_UPLOAD_PATH}",
            expiration_seconds=expiration_seconds,
            md5_checksum=obj.get_build().upload.md5_checksum)
        exp = (datetime.datetime.now(datetime.timezone.utc) +
               datetime.timedelta(seconds=expiration_seconds)
               ).strftime("%Y-%m-%dT%H:%M:%SZ")
        return resp.url, exp

    # -- kaniko job construction --------------------------------------------
    def _base_job(self, obj, build_args: list[str], init_containers: list,
                  volumes: list, volume_mounts: list) -> dict:
        image = self.cloud.object_built_image_url(obj)
        return {
            "apiVersion": "batch/v1", "kind": "Job",
            "metadata": {
                "name": build_job_name(obj),
                "namespace": obj.namespace,
                "annotations": {"image": image},
                "ownerReferences": [_owner_ref(obj)],
            },
            "spec": {
                "backoffLimit": 1,
                "template": {
                    "metadata": {
                        "annotations": {
                            "kubectl.kubernetes.io/default-container":
                                "builder"},
                        "labels": {self.kind.lower(): obj.name,
                                   "role": "build"},
                    },
                    "spec": {
                        "initContainers": init_containers,
                        "securityContext": {"runAsUser": 0, "runAsGroup": 0,
                                            "fsGroup": 3003},
                        "serviceAccountName": SA_CONTAINER_BUILDER,
                        "containers": [{
                            "name": "builder",
                            "image": BUILDER_IMAGE,
                            "args": build_args,
                            "volumeMounts": volume_mounts,
                            "resources": container_builder_resources(
                                self.cloud.name),
                        }],
                        "restartPolicy": "Never",
                        "volumes": volumes,
                    },
                },
            },
        }

    def _common_build_args(self, context: str, obj) -> list[str]:
        return [
            f"--context={context}",
            f"--destination={self.cloud.object_built_image_url(obj)}",
            "--cache=true",
            "--compressed-caching=false",
            "--log-format=color",
            "--log-timestamp=false",
        ]

    def _git_build_job(self, obj) -> dict:
        git = obj.get_build().git
        build_args = self._common_build_args("dir:///workspace", obj)
        if git.path:
            build_args.append(f"--context-sub-path={git.path}")
        clone_args = ["clone", git.url]
        if git.tag:
            clone_args += ["--branch", git.tag]
        elif git.branch:
            clone_args += ["--branch", git.branch]
        clone_args.append("/workspace")
        ws_mount = [{"name": "workspace", "mountPath": "/workspace"}]
        init = [{"name": "git-clone", "image": GIT_IMAGE,
                 "args": clone_args, "volumeMounts": list(ws_mount)}]
        volumes = [{"name": "workspace", "emptyDir": {}}]
        return self._base_job(obj, build_args, init, volumes, ws_mount)

    def _storage_build_job(self, obj) -> dict:
        context = (f"{self.cloud.object_artifact_url(obj)}/"
                   f"{LATEST_UPLOAD_PATH}")
        build_args = self._common_build_args(context, obj)
        mounts = [{"name": "workspace", "mountPath": "/workspace"}]
        volumes = [{"name": "workspace", "emptyDir": {}}]
        if self.cloud.name == "kind":
            # tar:// context resolved through the hostPath bucket
            # (reference build_reconciler.go:453-468).
            mounts.append({"name": "bucket", "mountPath": "/bucket"})
            volumes.append({"name": "bucket", "hostPath": {
                "path": "/bucket", "type": "Directory"}})
        return self._base_job(obj, build_args, [], volumes, mounts)


def _owner_ref(obj) -> dict:
    return {
        "apiVersion": "substratus.ai/v1", "kind": obj.kind,
        "name": obj.name, "uid": obj.metadata.get("uid", ""),
        "controller": True, "blockOwnerDeletion": True,
    }
