"""Manifest discovery: find + decode YAML, keep substratus kinds.

Parity: reference internal/tui/manifests.go:130-262.
"""
from __future__ import annotations

import os
import re

import yaml

from ..api.types import object_from_manifest


def find_manifests(path: str, kind_filter: str = "") -> list:
    """Return typed substratus objects from `path` (file or directory)."""
    files = []
    if os.path.isfile(path):
        files = [path]
    else:
        for name in sorted(os.listdir(path)):
            if name.endswith((".yaml", ".yml")):
                files.append(os.path.join(path, name))
    out = []
    for f in files:
        with open(f) as fh:
            try:
                docs = list(yaml.safe_load_all(fh))
            except yaml.YAMLError:
                continue
        for d in docs:
            obj = object_from_manifest(d)
            if obj is None:
                continue
            if kind_filter and obj.kind.lower() != kind_filter.lower():
                continue
            obj._source_file = f
            out.append(obj)
    return out


_VERSION_RE = re.compile(r"^(?P<base>.*?)-(?P<n>\d+)$")


def next_version_name(kube, kind: str, namespace: str, base: str) -> str:
    """Auto-versioning `name-N` used by `sub run -i`
    (reference internal/tui/common.go:158-265): scan existing objects
    whose names are `base` or `base-N`, return base-(maxN+1)."""
    m = _VERSION_RE.match(base)
    if m:
        base = m.group("base")
    max_n = 0
    for o in kube.list("substratus.ai/v1", kind, namespace):
        name = o["metadata"]["name"]
        if name == base:
            max_n = max(max_n, 1)
            continue
        mm = _VERSION_RE.match(name)
        if mm and mm.group("base") == base:
            max_n = max(max_n, int(mm.group("n")) + 1)
    return f"{base}-{max_n}" if max_n else base
