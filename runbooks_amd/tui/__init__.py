"""Terminal UI for the `sub` flows.

Parity: reference internal/tui/ (bubbletea sub-models composed per
command — manifests picker `manifests.go:130-262`, upload progress
`upload.go:16-170`, readiness checklist `readiness.go:15-100`, pod log
view `pods.go:23-231`). Rendered with rich (live tables + progress bars);
selection prompts fall back to first-match when not a TTY so scripted
use keeps working.
"""
from .views import (
    ReadinessChecklist,
    UploadProgress,
    format_conditions,
    select_manifest,
)

__all__ = ["select_manifest", "UploadProgress", "ReadinessChecklist",
           "format_conditions"]
