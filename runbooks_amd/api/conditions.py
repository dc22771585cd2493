"""Condition types and reasons (parity: reference api/v1/conditions.go:3-32)."""

CONDITION_UPLOADED = "Uploaded"
CONDITION_BUILT = "Built"
CONDITION_COMPLETE = "Complete"
CONDITION_SERVING = "Serving"

REASON_MODEL_NOT_FOUND = "ModelNotFound"
REASON_MODEL_NOT_READY = "ModelNotReady"

REASON_BASE_MODEL_NOT_FOUND = "BaseModelNotFound"
REASON_BASE_MODEL_NOT_READY = "BaseModelNotReady"

REASON_DATASET_NOT_FOUND = "DatasetNotFound"
REASON_DATASET_NOT_READY = "ReasonDatasetNotReady"

REASON_JOB_NOT_COMPLETE = "JobNotComplete"
REASON_JOB_COMPLETE = "JobComplete"
REASON_JOB_FAILED = "JobFailed"
REASON_DEPLOYMENT_READY = "DeploymentReady"
REASON_DEPLOYMENT_NOT_READY = "DeploymentNotReady"
REASON_POD_READY = "PodReady"
REASON_POD_NOT_READY = "PodNotReady"

REASON_SUSPENDED = "Suspended"

REASON_AWAITING_UPLOAD = "AwaitingUpload"
REASON_UPLOAD_FOUND = "UploadFound"
