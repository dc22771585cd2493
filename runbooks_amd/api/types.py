"""substratus.ai/v1 custom-resource types.

Field-for-field compatible with the reference CRD schemas so its example
manifests apply unchanged:

- shared types   reference api/v1/common_types.go:8-111
- Model          reference api/v1/model_types.go:10-36 (spec), :86-103 (status)
- Dataset        reference api/v1/dataset_types.go:10-28
- Server         reference api/v1/server_types.go:10-31
- Notebook       reference api/v1/notebook_types.go:10-38

The one deliberate divergence: GPUType gains AMD Instinct values and the
default accelerator is amd-mi355x scheduled as `amd.com/gpu` via the ROCm
k8s-device-plugin (reference common_types.go:94-107 lists only NVIDIA types).

Objects are dataclasses with lossless ``to_dict``/``from_dict`` matching the
reference's JSON wire shape, so they round-trip through the K8s API (real or
the in-memory test server) byte-identically.
"""
from __future__ import annotations

import copy
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Optional, Union

GROUP = "substratus.ai"
VERSION = "v1"
API_VERSION = f"{GROUP}/{VERSION}"

# .spec.params values are int-or-string (reference model_types.go:35 uses
# intstr.IntOrString); booleans appear in example manifests too.
ParamValue = Union[int, str, bool]


def _drop_none(d: dict) -> dict:
    return {k: v for k, v in d.items() if v is not None and v != {} and v != []}


def now_rfc3339() -> str:
    return datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")


# --------------------------------------------------------------------------
# Shared spec types (reference api/v1/common_types.go)
# --------------------------------------------------------------------------

@dataclass
class BuildGit:
    """Git build source (reference common_types.go:32-48)."""
    url: str = ""
    path: str = ""
    tag: str = ""
    branch: str = ""

    def to_dict(self) -> dict:
        return _drop_none({"url": self.url, "path": self.path or None,
                           "tag": self.tag or None, "branch": self.branch or None})

    @classmethod
    def from_dict(cls, d: dict) -> "BuildGit":
        return cls(url=d.get("url", ""), path=d.get("path", ""),
                   tag=d.get("tag", ""), branch=d.get("branch", ""))


@dataclass
class BuildUpload:
    """Client-upload build source (reference common_types.go:17-30)."""
    md5_checksum: str = ""
    request_id: str = ""

    def to_dict(self) -> dict:
        return {"md5Checksum": self.md5_checksum, "requestID": self.request_id}

    @classmethod
    def from_dict(cls, d: dict) -> "BuildUpload":
        return cls(md5_checksum=d.get("md5Checksum", ""),
                   request_id=d.get("requestID", ""))


@dataclass
class Build:
    """Image build request: git xor upload (reference common_types.go:8-15)."""
    git: Optional[BuildGit] = None
    upload: Optional[BuildUpload] = None

    def to_dict(self) -> dict:
        return _drop_none({
            "git": self.git.to_dict() if self.git else None,
            "upload": self.upload.to_dict() if self.upload else None,
        })

    @classmethod
    def from_dict(cls, d: dict) -> "Build":
        return cls(
            git=BuildGit.from_dict(d["git"]) if d.get("git") else None,
            upload=BuildUpload.from_dict(d["upload"]) if d.get("upload") else None,
        )


@dataclass
class UploadStatus:
    """Signed-URL handshake state (reference common_types.go:51-71)."""
    signed_url: str = ""
    request_id: str = ""
    expiration: str = ""          # RFC3339
    stored_md5_checksum: str = ""

    def to_dict(self) -> dict:
        return _drop_none({
            "signedURL": self.signed_url or None,
            "requestID": self.request_id or None,
            "expiration": self.expiration or None,
            "storedMD5Checksum": self.stored_md5_checksum or None,
        })

    @classmethod
    def from_dict(cls, d: dict) -> "UploadStatus":
        return cls(signed_url=d.get("signedURL", ""),
                   request_id=d.get("requestID", ""),
                   expiration=d.get("expiration", ""),
                   stored_md5_checksum=d.get("storedMD5Checksum", ""))


@dataclass
class ObjectRef:
    """Same-namespace object reference (reference common_types.go:74-79)."""
    name: str = ""

    def to_dict(self) -> dict:
        return {"name": self.name}

    @classmethod
    def from_dict(cls, d: dict) -> "ObjectRef":
        return cls(name=d.get("name", ""))


class GPUType(str):
    """Accelerator model (reference common_types.go:94-100 + AMD values)."""


GPU_AMD_MI355X = GPUType("amd-mi355x")
GPU_AMD_MI300X = GPUType("amd-mi300x")
# Accepted for manifest compatibility with the reference; mapped onto the
# MI355X pool by the resources table (runbooks_amd/resources.py).
GPU_NVIDIA_A100 = GPUType("nvidia-a100")
GPU_NVIDIA_T4 = GPUType("nvidia-t4")
GPU_NVIDIA_L4 = GPUType("nvidia-l4")


@dataclass
class GPUResources:
    """(reference common_types.go:102-107)"""
    type: str = ""
    count: int = 0

    def to_dict(self) -> dict:
        return _drop_none({"type": self.type or None,
                           "count": self.count or None})

    @classmethod
    def from_dict(cls, d: dict) -> "GPUResources":
        return cls(type=d.get("type", ""), count=int(d.get("count", 0) or 0))


@dataclass
class Resources:
    """Compute resources with the reference's defaults
    (reference common_types.go:81-92: cpu=2, disk=10, memory=10)."""
    cpu: int = 2
    disk: int = 10
    memory: int = 10
    gpu: Optional[GPUResources] = None

    def to_dict(self) -> dict:
        d = {"cpu": self.cpu, "disk": self.disk, "memory": self.memory}
        if self.gpu:
            d["gpu"] = self.gpu.to_dict()
        return d

    @classmethod
    def from_dict(cls, d: dict) -> "Resources":
        return cls(cpu=int(d.get("cpu", 2) or 2),
                   disk=int(d.get("disk", 10) or 10),
                   memory=int(d.get("memory", 10) or 10),
                   gpu=GPUResources.from_dict(d["gpu"]) if d.get("gpu") else None)


@dataclass
class ArtifactsStatus:
    """(reference common_types.go:109-111)"""
    url: str = ""

    def to_dict(self) -> dict:
        return _drop_none({"url": self.url or None})

    @classmethod
    def from_dict(cls, d: dict) -> "ArtifactsStatus":
        return cls(url=d.get("url", ""))


# --------------------------------------------------------------------------
# Object base — metadata + conditions plumbing shared by all four kinds
# --------------------------------------------------------------------------

@dataclass
class _Base:
    name: str = ""
    namespace: str = "default"
    metadata: dict = field(default_factory=dict)   # labels/annotations/uid/...

    command: list[str] = field(default_factory=list)
    env: dict[str, str] = field(default_factory=dict)
    image: Optional[str] = None
    build: Optional[Build] = None
    resources: Optional[Resources] = None
    params: dict[str, ParamValue] = field(default_factory=dict)

    ready: bool = False
    conditions: list[dict] = field(default_factory=list)
    artifacts: ArtifactsStatus = field(default_factory=ArtifactsStatus)
    build_upload: UploadStatus = field(default_factory=UploadStatus)

    kind = ""  # overridden

    # -- accessors matching the reference's BuildableObject interface
    #    (reference internal/controller/build_reconciler.go:31-42)
    def get_image(self) -> str:
        return self.image or ""

    def set_image(self, image: str) -> None:
        self.image = image

    def get_build(self) -> Optional[Build]:
        return self.build

    def set_build(self, b: Optional[Build]) -> None:
        self.build = b

    def get_params(self) -> dict[str, ParamValue]:
        return self.params

    def get_status_ready(self) -> bool:
        return self.ready

    def set_status_ready(self, r: bool) -> None:
        self.ready = r

    def get_status_upload(self) -> UploadStatus:
        return self.build_upload

    def set_status_upload(self, u: UploadStatus) -> None:
        self.build_upload = u

    def get_status_artifacts(self) -> ArtifactsStatus:
        return self.artifacts

    # -- condition helpers (metav1.SetStatusCondition semantics)
    def set_condition(self, type_: str, status: bool, reason: str,
                      message: str = "", observed_generation: int = 0) -> None:
        cond = {
            "type": type_,
            "status": "True" if status else "False",
            "reason": reason,
            "message": message,
            "observedGeneration": observed_generation,
            "lastTransitionTime": now_rfc3339(),
        }
        for i, c in enumerate(self.conditions):
            if c["type"] == type_:
                if c["status"] == cond["status"]:
                    cond["lastTransitionTime"] = c.get("lastTransitionTime",
                                                       cond["lastTransitionTime"])
                self.conditions[i] = cond
                return
        self.conditions.append(cond)

    def get_condition(self, type_: str) -> Optional[dict]:
        for c in self.conditions:
            if c["type"] == type_:
                return c
        return None

    def is_condition_true(self, type_: str) -> bool:
        c = self.get_condition(type_)
        return bool(c) and c["status"] == "True"

    # -- serialization
    def _spec_dict(self) -> dict:
        return _drop_none({
            "command": list(self.command) or None,
            "env": dict(self.env) or None,
            "image": self.image,
            "build": self.build.to_dict() if self.build else None,
            "resources": self.resources.to_dict() if self.resources else None,
            "params": dict(self.params) or None,
        })

    def _status_dict(self) -> dict:
        d: dict[str, Any] = {"ready": self.ready}
        if self.conditions:
            d["conditions"] = copy.deepcopy(self.conditions)
        art = self.artifacts.to_dict()
        if art:
            d["artifacts"] = art
        up = self.build_upload.to_dict()
        if up:
            d["buildUpload"] = up
        return d

    def to_dict(self) -> dict:
        meta = dict(self.metadata)
        meta["name"] = self.name
        meta["namespace"] = self.namespace
        return {
            "apiVersion": API_VERSION,
            "kind": self.kind,
            "metadata": meta,
            "spec": self._spec_dict(),
            "status": self._status_dict(),
        }

    def _load_common(self, d: dict) -> None:
        meta = dict(d.get("metadata") or {})
        self.name = meta.pop("name", "")
        self.namespace = meta.pop("namespace", "default") or "default"
        self.metadata = meta
        spec = d.get("spec") or {}
        self.command = list(spec.get("command") or [])
        self.env = dict(spec.get("env") or {})
        self.image = spec.get("image")
        self.build = Build.from_dict(spec["build"]) if spec.get("build") else None
        self.resources = (Resources.from_dict(spec["resources"])
                          if spec.get("resources") else None)
        self.params = dict(spec.get("params") or {})
        status = d.get("status") or {}
        self.ready = bool(status.get("ready", False))
        self.conditions = copy.deepcopy(status.get("conditions") or [])
        self.artifacts = ArtifactsStatus.from_dict(status.get("artifacts") or {})
        self.build_upload = UploadStatus.from_dict(status.get("buildUpload") or {})

    @classmethod
    def from_dict(cls, d: dict):
        o = cls()
        o._load_common(d)
        return o

    @property
    def generation(self) -> int:
        return int(self.metadata.get("generation", 0) or 0)


# --------------------------------------------------------------------------
# The four kinds
# --------------------------------------------------------------------------

@dataclass
class Model(_Base):
    """Build/import/fine-tune job (reference api/v1/model_types.go:10-36)."""
    kind = "Model"
    model: Optional[ObjectRef] = None     # base model for transfer learning
    dataset: Optional[ObjectRef] = None

    def _spec_dict(self) -> dict:
        d = super()._spec_dict()
        if self.model:
            d["model"] = self.model.to_dict()
        if self.dataset:
            d["dataset"] = self.dataset.to_dict()
        return d

    @classmethod
    def from_dict(cls, d: dict) -> "Model":
        o = cls()
        o._load_common(d)
        spec = d.get("spec") or {}
        if spec.get("model"):
            o.model = ObjectRef.from_dict(spec["model"])
        elif spec.get("modelName"):
            # legacy flat form used by two of the reference's own
            # examples (facebook-opt-125m/finetuned-{notebook,server});
            # the CRD field is model: ObjectRef (server_types.go:27)
            o.model = ObjectRef(str(spec["modelName"]))
        if spec.get("dataset"):
            o.dataset = ObjectRef.from_dict(spec["dataset"])
        elif spec.get("datasetName"):
            o.dataset = ObjectRef(str(spec["datasetName"]))
        return o


@dataclass
class Dataset(_Base):
    """Containerized data loader (reference api/v1/dataset_types.go:10-28)."""
    kind = "Dataset"


@dataclass
class Server(_Base):
    """HTTP model server (reference api/v1/server_types.go:10-31)."""
    kind = "Server"
    model: ObjectRef = field(default_factory=ObjectRef)

    def _spec_dict(self) -> dict:
        d = super()._spec_dict()
        if self.model.name:
            d["model"] = self.model.to_dict()
        return d

    @classmethod
    def from_dict(cls, d: dict) -> "Server":
        o = cls()
        o._load_common(d)
        spec = d.get("spec") or {}
        if spec.get("model"):
            o.model = ObjectRef.from_dict(spec["model"])
        elif spec.get("modelName"):
            o.model = ObjectRef(str(spec["modelName"]))
        return o


@dataclass
class Notebook(_Base):
    """Jupyter pod (reference api/v1/notebook_types.go:10-38)."""
    kind = "Notebook"
    suspend: Optional[bool] = None
    model: Optional[ObjectRef] = None
    dataset: Optional[ObjectRef] = None

    def is_suspended(self) -> bool:
        """(reference notebook_types.go:87-89)"""
        return bool(self.suspend)

    def _spec_dict(self) -> dict:
        d = super()._spec_dict()
        if self.suspend is not None:
            d["suspend"] = self.suspend
        if self.model:
            d["model"] = self.model.to_dict()
        if self.dataset:
            d["dataset"] = self.dataset.to_dict()
        return d

    @classmethod
    def from_dict(cls, d: dict) -> "Notebook":
        o = cls()
        o._load_common(d)
        spec = d.get("spec") or {}
        o.suspend = spec.get("suspend")
        if spec.get("model"):
            o.model = ObjectRef.from_dict(spec["model"])
        elif spec.get("modelName"):
            # legacy flat form used by two of the reference's own
            # examples (facebook-opt-125m/finetuned-{notebook,server});
            # the CRD field is model: ObjectRef (server_types.go:27)
            o.model = ObjectRef(str(spec["modelName"]))
        if spec.get("dataset"):
            o.dataset = ObjectRef.from_dict(spec["dataset"])
        elif spec.get("datasetName"):
            o.dataset = ObjectRef(str(spec["datasetName"]))
        return o


KINDS: dict[str, type] = {
    "Model": Model, "Dataset": Dataset, "Server": Server, "Notebook": Notebook,
}
# Plural resource names used in API paths / CRDs.
PLURALS: dict[str, str] = {
    "Model": "models", "Dataset": "datasets",
    "Server": "servers", "Notebook": "notebooks",
}


def object_from_manifest(d: dict):
    """Decode one YAML/JSON manifest document into a typed object.

    Returns None for non-substratus kinds (the CLI's manifest scanner skips
    them, mirroring reference internal/tui/manifests.go:130-262).
    """
    if not isinstance(d, dict):
        return None
    if d.get("apiVersion") != API_VERSION:
        return None
    cls = KINDS.get(d.get("kind", ""))
    return cls.from_dict(d) if cls else None
