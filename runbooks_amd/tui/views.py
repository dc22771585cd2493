"""rich-backed view components for the CLI flows."""
from __future__ import annotations

import sys
from typing import Optional

try:
    from rich.console import Console
    from rich.live import Live
    from rich.progress import (
        BarColumn,
        Progress,
        TaskProgressColumn,
        TextColumn,
    )
    from rich.table import Table
    _HAVE_RICH = True
    _console = Console()
except ImportError:  # pragma: no cover - rich ships in the image
    _HAVE_RICH = False


def select_manifest(objs: list, assume_first: bool = False):
    """Pick one substratus object from discovered manifests — the
    interactive arrow-key picker (tui/core.py event loop, reference
    manifests.go) on a TTY; first match otherwise."""
    if len(objs) == 1 or assume_first or not sys.stdin.isatty():
        return objs[0]
    try:
        from .core import Program, SelectModel
        items = [f"{o.kind}/{o.name}  ({getattr(o, '_source_file', '?')})"
                 for o in objs]
        model = Program(SelectModel("select a manifest:", items)).run()
        if model.chosen is None:
            raise SystemExit(1)
        return objs[model.chosen]
    except (ImportError, OSError):  # no termios (exotic terminal): prompt
        pass
    for i, o in enumerate(objs):
        src = getattr(o, "_source_file", "?")
        print(f"  [{i}] {o.kind}/{o.name}  ({src})")
    while True:
        raw = input(f"select manifest [0-{len(objs) - 1}]: ").strip()
        try:
            idx = int(raw)
            if 0 <= idx < len(objs):
                return objs[idx]
        except ValueError:
            pass


class UploadProgress:
    """Progress bar over tarball build + upload
    (reference tui/upload.go:16-170)."""

    def __init__(self, label: str = "upload"):
        self._progress = None
        self._task = None
        self.label = label
        self.files: list[str] = []

    def __enter__(self):
        if _HAVE_RICH:
            self._progress = Progress(
                TextColumn("[bold]{task.description}"), BarColumn(),
                TaskProgressColumn(), console=_console, transient=True)
            self._progress.start()
            self._task = self._progress.add_task(self.label, total=1.0)
        return self

    def __exit__(self, *exc):
        if self._progress:
            self._progress.stop()
        return False

    def on_file(self, path: str):
        self.files.append(path)
        if self._progress:
            self._progress.update(self._task,
                                  description=f"{self.label}: {path[-40:]}")

    def on_fraction(self, frac: float):
        if self._progress:
            self._progress.update(self._task, completed=frac)


def format_conditions(raw: dict) -> list[tuple[str, str, str]]:
    """(mark, type, reason) rows for an object's conditions checklist —
    pure so the view is unit-testable (reference readiness.go:15-100)."""
    rows = []
    status = (raw or {}).get("status") or {}
    for c in status.get("conditions") or []:
        mark = "✓" if c.get("status") == "True" else "…"
        rows.append((mark, c.get("type", ""), c.get("reason", "")))
    if status.get("ready"):
        rows.append(("✓", "Ready", ""))
    else:
        rows.append(("…", "Ready", ""))
    return rows


class ReadinessChecklist:
    """Live conditions table driven by wait_ready's callback."""

    def __init__(self, title: str):
        self.title = title
        self._live: Optional[Live] = None
        self._last = None

    def __enter__(self):
        if _HAVE_RICH:
            self._live = Live(console=_console, refresh_per_second=4,
                              transient=True)
            self._live.start()
        return self

    def __exit__(self, *exc):
        if self._live:
            self._live.stop()
        return False

    def update(self, raw: dict):
        rows = format_conditions(raw)
        if rows == self._last:
            return
        self._last = rows
        if self._live:
            t = Table(title=self.title, show_header=False, box=None)
            for mark, typ, reason in rows:
                style = "green" if mark == "✓" else "yellow"
                t.add_row(f"[{style}]{mark}[/{style}]", typ, reason)
            self._live.update(t)
        else:  # plain fallback
            print("; ".join(f"{m} {t}({r})" for m, t, r in rows))
