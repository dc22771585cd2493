"""substratus.ai/v1 API types (MI355X edition).

API-compatible with the reference CRDs (reference api/v1/*.go) so that the
reference's example manifests apply unchanged, with the GPUType values
extended for AMD Instinct accelerators (amd-mi355x et al.).
"""
from .conditions import (
    CONDITION_BUILT,
    CONDITION_COMPLETE,
    CONDITION_SERVING,
    CONDITION_UPLOADED,
    REASON_AWAITING_UPLOAD,
    REASON_BASE_MODEL_NOT_FOUND,
    REASON_BASE_MODEL_NOT_READY,
    REASON_DATASET_NOT_FOUND,
    REASON_DATASET_NOT_READY,
    REASON_DEPLOYMENT_NOT_READY,
    REASON_DEPLOYMENT_READY,
    REASON_JOB_COMPLETE,
    REASON_JOB_FAILED,
    REASON_JOB_NOT_COMPLETE,
    REASON_MODEL_NOT_FOUND,
    REASON_MODEL_NOT_READY,
    REASON_POD_NOT_READY,
    REASON_POD_READY,
    REASON_SUSPENDED,
    REASON_UPLOAD_FOUND,
)
from .types import (
    GROUP,
    VERSION,
    ArtifactsStatus,
    Build,
    BuildGit,
    BuildUpload,
    Dataset,
    GPUResources,
    GPUType,
    Model,
    Notebook,
    ObjectRef,
    Resources,
    Server,
    UploadStatus,
    object_from_manifest,
)

__all__ = [
    "GROUP", "VERSION",
    "Model", "Dataset", "Server", "Notebook",
    "Build", "BuildGit", "BuildUpload", "UploadStatus", "ObjectRef",
    "Resources", "GPUResources", "GPUType", "ArtifactsStatus",
    "object_from_manifest",
    "CONDITION_UPLOADED", "CONDITION_BUILT", "CONDITION_COMPLETE",
    "CONDITION_SERVING",
    "REASON_MODEL_NOT_FOUND", "REASON_MODEL_NOT_READY",
    "REASON_BASE_MODEL_NOT_FOUND", "REASON_BASE_MODEL_NOT_READY",
    "REASON_DATASET_NOT_FOUND", "REASON_DATASET_NOT_READY",
    "REASON_JOB_NOT_COMPLETE", "REASON_JOB_COMPLETE", "REASON_JOB_FAILED",
    "REASON_DEPLOYMENT_READY", "REASON_DEPLOYMENT_NOT_READY",
    "REASON_POD_READY", "REASON_POD_NOT_READY",
    "REASON_SUSPENDED", "REASON_AWAITING_UPLOAD", "REASON_UPLOAD_FOUND",
]
