"""Minimal Kubernetes client layer.

The reference uses controller-runtime's client (Go). Here the same role is
filled by a small dynamic client interface (`KubeClient`) with two
implementations: `HTTPKubeClient` (real API server over HTTPS, used by the
deployed controller-manager) and `MemoryKubeClient` (in-memory API server
semantics — create/apply/patch/status/watch/resourceVersion — used by the
integration tests exactly the way the reference uses envtest,
reference internal/controller/main_test.go:46-191).
"""
from .client import HTTPKubeClient, KubeClient, NotFound, Conflict
from .memory import MemoryKubeClient

__all__ = ["KubeClient", "HTTPKubeClient", "MemoryKubeClient", "NotFound",
           "Conflict"]
