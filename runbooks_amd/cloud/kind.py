"""Local kind-cluster cloud (parity: reference internal/cloud/kind.go).

Bucket is a hostPath `/bucket` on the kind node exposed as `tar:///bucket`
(kind.go:23-48); registry discovered from the in-cluster registry Service's
injected env var (kind.go:14-17); identity is a no-op (kind.go:92-94).
"""
from __future__ import annotations

from .base import BucketURL, Cloud, MountBucketConfig


class Kind(Cloud):
    name = "kind"

    def __init__(self, env=None):
        super().__init__(env)
        self.registry_discovery_ip = self._env.get(
            "REGISTRY_PORT_5000_TCP_ADDR", "")

    def auto_configure(self) -> None:
        if self.artifact_bucket_url is None:
            # "tar:///bucket" — hostPath /bucket on the kind node
            # (reference kind.go:23-48); path stored without the leading
            # slash, as ParseBucketURL does.
            self.artifact_bucket_url = BucketURL(scheme="tar", bucket="",
                                                 path="bucket")
        if not self.registry_url and self.registry_discovery_ip:
            self.registry_url = f"{self.registry_discovery_ip}:5000"

    def mount_bucket(self, pod_metadata: dict, pod_spec: dict, obj,
                     req: MountBucketConfig) -> None:
        """hostPath volume + SubPath mounts under /content
        (reference kind.go:50-90)."""
        bkt = self._artifact_bucket_for(obj)
        pod_spec.setdefault("volumes", []).append({
            "name": req.name,
            "hostPath": {"path": "/" + bkt.path.lstrip("/"),
                         "type": "DirectoryOrCreate"},
        })
        self._attach_mounts(pod_spec, req, lambda m: m.bucket_subdir)

    def associate_principal(self, sa: dict) -> None:
        pass

    def get_principal(self, sa: dict) -> tuple[str, bool]:
        return "", True
