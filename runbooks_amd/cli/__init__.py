"""`sub` CLI — kubectl-plugin style front end.

Parity: reference cmd/sub/main.go + internal/cli/ (cobra commands
apply/notebook/run/serve/get/delete, root.go:9-24) and the bubbletea TUI
orchestration in internal/tui/ (manifest discovery, upload progress,
readiness checklists, auto-versioned `run -i`). Rendering uses rich
instead of bubbletea; the flows and flags match.
"""
from .main import main

__all__ = ["main"]
