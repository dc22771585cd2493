"""Build-context tarball + the signed-URL upload flow.

Parity: reference internal/client/upload.go —
- PrepareImageTarball (:38-68): Dockerfile required, tar.gz of regular
  files/dirs only, md5 of the archive
- SetUploadContainerSpec (:70-93): spec.build.upload = {md5, requestID}
- Upload (:126-192): watch status.buildUpload until the signed URL for our
  requestID appears (or a stored matching md5 short-circuits), HTTP PUT
  with Content-MD5, then patch an `upload-timestamp` annotation to requeue
- WaitReady (client.go:114-135): 1 s status.ready poll
"""
from __future__ import annotations

import base64
import hashlib
import os
import tarfile
import tempfile
import time
import urllib.request
from dataclasses import dataclass
from typing import Callable, Optional

from ..api.types import Build, BuildUpload
from ..k8s import KubeClient


@dataclass
class Tarball:
    temp_dir: str
    path: str
    md5_checksum: str


def prepare_image_tarball(build_path: str,
                          progress: Optional[Callable[[str], None]] = None
                          ) -> Tarball:
    if not os.path.isfile(os.path.join(build_path, "Dockerfile")):
        raise FileNotFoundError(
            f"path does not contain Dockerfile: {build_path}")
    tmp = tempfile.mkdtemp(prefix="runbooks-upload")
    tar_path = os.path.join(tmp, "archive.tar.gz")
    with tarfile.open(tar_path, "w:gz") as tf:
        for root, dirs, files in os.walk(build_path):
            for name in sorted(dirs) + sorted(files):
                p = os.path.join(root, name)
                if not (os.path.isdir(p) or os.path.isfile(p)):
                    continue  # regular files and dirs only
                rel = os.path.relpath(p, build_path)
                tf.add(p, arcname=rel, recursive=False)
                if progress:
                    progress(p)
    md5 = hashlib.md5()
    with open(tar_path, "rb") as f:
        for chunk in iter(lambda: f.read(1 << 20), b""):
            md5.update(chunk)
    return Tarball(temp_dir=tmp, path=tar_path, md5_checksum=md5.hexdigest())


def set_upload_container_spec(obj, tb: Tarball, request_id: str) -> None:
    b = obj.get_build() or Build()
    b.git = None
    b.upload = BuildUpload(md5_checksum=tb.md5_checksum,
                           request_id=request_id)
    obj.set_build(b)


def clear_image(obj) -> None:
    obj.set_image("")
    obj.image = None


def upload(kube: KubeClient, obj, tb: Tarball,
           progress: Optional[Callable[[float], None]] = None,
           timeout: float = 300.0) -> None:
    """Wait for the controller's signed URL, PUT the tarball, then patch
    the upload-timestamp annotation so the controller requeues."""
    cls = type(obj)
    deadline = time.time() + timeout
    url = None
    while time.time() < deadline:
        raw = kube.get("substratus.ai/v1", obj.kind, obj.namespace, obj.name)
        if raw is None:
            raise RuntimeError(f"{obj.kind}/{obj.name} deleted during upload")
        cur = cls.from_dict(raw)
        status = cur.get_status_upload()
        spec = cur.get_build().upload if cur.get_build() else None
        if spec is None:
            raise RuntimeError("object has no upload build spec")
        if status.stored_md5_checksum == tb.md5_checksum:
            return  # already in storage (controller matched the checksum)
        if status.signed_url and status.request_id == spec.request_id:
            url = status.signed_url
            break
        time.sleep(0.5)
    if url is None:
        raise TimeoutError("timed out waiting for signed upload URL")

    with open(tb.path, "rb") as f:
        body = f.read()
    req = urllib.request.Request(url, data=body, method="PUT", headers={
        "Content-Type": "application/octet-stream",
        "Content-MD5":
            base64.b64encode(bytes.fromhex(tb.md5_checksum)).decode(),
    })
    with urllib.request.urlopen(req) as resp:
        if resp.status not in (200, 201, 204):
            raise RuntimeError(f"upload failed: HTTP {resp.status}")
    if progress:
        progress(1.0)

    ts = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
    kube.patch("substratus.ai/v1", obj.kind, obj.namespace, obj.name,
               {"metadata": {"annotations": {"upload-timestamp": ts}}})


def wait_ready(kube: KubeClient, obj, timeout: float = 1800.0,
               interval: float = 1.0,
               callback: Optional[Callable[[dict], None]] = None) -> dict:
    """Poll status.ready (reference client.go:114-135)."""
    deadline = time.time() + timeout
    while time.time() < deadline:
        raw = kube.get("substratus.ai/v1", obj.kind, obj.namespace, obj.name)
        if raw is not None:
            if callback:
                callback(raw)
            if (raw.get("status") or {}).get("ready"):
                return raw
        time.sleep(interval)
    raise TimeoutError(f"{obj.kind}/{obj.name} not ready after {timeout}s")
