"""GCP/GKE cloud (parity: reference internal/cloud/gcp.go).

GCS FUSE CSI bucket mounts with the gke-gcsfuse pod annotations
(gcp.go:73-124) and workload-identity principal association on service
accounts (gcp.go:126-140). Autoconfiguration uses explicit env vars
(PROJECT_ID, CLUSTER_LOCATION) — the reference additionally probes the GCE
metadata server (gcp.go:28-56), which does not exist off-GCE.
"""
from __future__ import annotations

from .base import BucketURL, Cloud, MountBucketConfig

WORKLOAD_IDENTITY_ANNOTATION = "iam.gke.io/gcp-service-account"


class GCP(Cloud):
    name = "gcp"

    def __init__(self, env=None):
        super().__init__(env)
        self.project_id = self._env.get("PROJECT_ID", "")
        self.cluster_location = self._env.get("CLUSTER_LOCATION", "")

    def region(self) -> str:
        # cluster location may be a zone ("us-central1-a") → region
        parts = self.cluster_location.split("-")
        return "-".join(parts[:2]) if len(parts) == 3 else self.cluster_location

    def auto_configure(self) -> None:
        if not self.registry_url and self.project_id:
            self.registry_url = (f"{self.region()}-docker.pkg.dev/"
                                 f"{self.project_id}/substratus")
        if self.artifact_bucket_url is None and self.project_id:
            self.artifact_bucket_url = BucketURL(
                scheme="gs", bucket=f"{self.project_id}-substratus-artifacts")
        if not self.principal and self.project_id:
            self.principal = \
                f"substratus@{self.project_id}.iam.gserviceaccount.com"

    def mount_bucket(self, pod_metadata: dict, pod_spec: dict, obj,
                     req: MountBucketConfig) -> None:
        ann = pod_metadata.setdefault("annotations", {})
        ann["gke-gcsfuse/volumes"] = "true"
        ann["gke-gcsfuse/cpu-limit"] = "2"
        ann["gke-gcsfuse/memory-limit"] = "800Mi"
        ann["gke-gcsfuse/ephemeral-storage-limit"] = "100Gi"

        bkt = self._artifact_bucket_for(obj)
        pod_spec.setdefault("volumes", []).append({
            "name": req.name,
            "csi": {
                "driver": "gcsfuse.csi.storage.gke.io",
                "readOnly": req.read_only,
                "volumeAttributes": {
                    "bucketName": bkt.bucket,
                    "mountOptions": "implicit-dirs,uid=0,gid=3003",
                },
            },
        })
        self._attach_mounts(
            pod_spec, req,
            lambda m: (bkt.path + "/" + m.bucket_subdir).lstrip("/"))

    def associate_principal(self, sa: dict) -> None:
        sa.setdefault("metadata", {}).setdefault("annotations", {})[
            WORKLOAD_IDENTITY_ANNOTATION] = self.principal

    def get_principal(self, sa: dict) -> tuple[str, bool]:
        ann = (sa.get("metadata") or {}).get("annotations") or {}
        bound = ann.get(WORKLOAD_IDENTITY_ANNOTATION) == self.principal
        return self.principal, bound
