"""Cloud abstraction (parity: reference internal/cloud/cloud.go:20-46).

`Cloud` hides registry naming, artifact-bucket naming/hashing, bucket
mounting into pod specs, and workload-identity association. Implementations:
`Kind` (local hostPath bucket + in-cluster registry) and `GCP` (GCS FUSE CSI
+ workload identity). The factory `new_cloud()` picks by $CLOUD.
"""
from .base import BucketURL, Cloud, Mount, MountBucketConfig, new_cloud, parse_bucket_url
from .gcp import GCP
from .kind import Kind

__all__ = ["Cloud", "Kind", "GCP", "BucketURL", "Mount", "MountBucketConfig",
           "new_cloud", "parse_bucket_url"]
