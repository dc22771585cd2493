"""Cloud interface, bucket URLs, naming/hashing.

Parity with the reference:
- Cloud interface          reference internal/cloud/cloud.go:20-46
- factory ($CLOUD env)     reference internal/cloud/cloud.go:48-85
- BucketURL parse/format   reference internal/cloud/utils.go
- image URL scheme         reference internal/cloud/common.go:18-43
  ({registry}/{cluster}-{kind}-{ns}-{name}:{tag}, tag from git tag/branch
  or upload md5, else "latest")
- artifact URL = bucket/md5("clusters/{c}/namespaces/{ns}/{kind}s/{name}")
                           reference internal/cloud/common.go:45-66
"""
from __future__ import annotations

import hashlib
import os
import urllib.parse
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class BucketURL:
    scheme: str = ""
    bucket: str = ""
    path: str = ""

    def __str__(self) -> str:
        return f"{self.scheme}://{self.bucket}/{self.path}"


def parse_bucket_url(s: str) -> BucketURL:
    u = urllib.parse.urlparse(s)
    # kind's "tar:///bucket" has an empty host
    return BucketURL(scheme=u.scheme, bucket=u.netloc,
                     path=u.path.lstrip("/"))


@dataclass
class Mount:
    bucket_subdir: str
    content_subdir: str


@dataclass
class MountBucketConfig:
    name: str                 # volume name
    container: str            # target container name
    mounts: list[Mount] = field(default_factory=list)
    read_only: bool = False


class CloudConfigError(Exception):
    pass


class Cloud:
    """Configured from env (CLUSTER_NAME, ARTIFACT_BUCKET_URL, REGISTRY_URL,
    PRINCIPAL — reference internal/cloud/common.go:11-16) with per-cloud
    autoconfiguration filling the gaps."""

    name = ""

    def __init__(self, env: Optional[dict] = None):
        env = dict(os.environ if env is None else env)
        self.cluster_name: str = env.get("CLUSTER_NAME", "")
        url = env.get("ARTIFACT_BUCKET_URL", "")
        self.artifact_bucket_url: Optional[BucketURL] = \
            parse_bucket_url(url) if url else None
        self.registry_url: str = env.get("REGISTRY_URL", "")
        self.principal: str = env.get("PRINCIPAL", "")
        self._env = env

    def auto_configure(self) -> None:
        raise NotImplementedError

    def validate(self) -> None:
        missing = [n for n, v in [("CLUSTER_NAME", self.cluster_name),
                                  ("ARTIFACT_BUCKET_URL",
                                   self.artifact_bucket_url),
                                  ("REGISTRY_URL", self.registry_url)]
                   if not v]
        if missing:
            raise CloudConfigError(f"missing cloud config: {missing}")

    # -- naming ------------------------------------------------------------
    def object_built_image_url(self, obj) -> str:
        if not obj.kind:
            raise ValueError("kind is empty")
        build = obj.get_build()
        tag = "latest"
        if build is not None:
            if build.git is not None:
                if build.git.tag:
                    tag = build.git.tag
                elif build.git.branch:
                    tag = build.git.branch
            elif build.upload is not None:
                tag = build.upload.md5_checksum
        return (f"{self.registry_url}/{self.cluster_name}-{obj.kind.lower()}"
                f"-{obj.namespace}-{obj.name}:{tag}")

    def object_artifact_url(self, obj) -> BucketURL:
        base = self.artifact_bucket_url
        assert base is not None, "cloud not configured"
        h = object_hash(self.cluster_name, obj)
        path = f"{base.path}/{h}" if base.path else h
        return BucketURL(scheme=base.scheme, bucket=base.bucket, path=path)

    # -- pod mutation / identity --------------------------------------------
    def mount_bucket(self, pod_metadata: dict, pod_spec: dict, obj,
                     req: MountBucketConfig) -> None:
        raise NotImplementedError

    def associate_principal(self, sa: dict) -> None:
        raise NotImplementedError

    def get_principal(self, sa: dict) -> tuple[str, bool]:
        """Returns (principal, bound?)."""
        raise NotImplementedError

    # -- shared mount helper -------------------------------------------------
    def _artifact_bucket_for(self, obj) -> BucketURL:
        status_url = obj.get_status_artifacts().url
        if status_url:
            return parse_bucket_url(status_url)
        return self.object_artifact_url(obj)

    @staticmethod
    def _attach_mounts(pod_spec: dict, req: MountBucketConfig,
                       subpath_of) -> None:
        for c in pod_spec.get("containers", []):
            if c["name"] == req.container:
                vm = c.setdefault("volumeMounts", [])
                for m in req.mounts:
                    vm.append({
                        "name": req.name,
                        "mountPath": "/content/" + m.content_subdir,
                        "subPath": subpath_of(m),
                        "readOnly": req.read_only,
                    })
                return
        raise ValueError(f"container not found: {req.container}")


def object_hash_input(cluster: str, obj) -> str:
    """(reference internal/cloud/common.go:58-66)"""
    if not obj.kind:
        raise ValueError("kind is empty")
    return (f"clusters/{cluster}/namespaces/{obj.namespace}/"
            f"{obj.kind.lower()}s/{obj.name}")


def object_hash(cluster: str, obj) -> str:
    return hashlib.md5(object_hash_input(cluster, obj).encode()).hexdigest()


def new_cloud(env: Optional[dict] = None) -> Cloud:
    """Factory: $CLOUD selects the implementation
    (reference internal/cloud/cloud.go:48-85; the GCE metadata probe is
    replaced by an explicit env var — there is no metadata server to probe
    in an air-gapped MI355X pod)."""
    from .gcp import GCP
    from .kind import Kind

    e = dict(os.environ if env is None else env)
    name = e.get("CLOUD", "")
    impls = {"kind": Kind, "gcp": GCP}
    if name not in impls:
        raise CloudConfigError(
            f"$CLOUD must be one of {sorted(impls)}, got {name!r}")
    cloud = impls[name](e)
    cloud.auto_configure()
    cloud.validate()
    return cloud
