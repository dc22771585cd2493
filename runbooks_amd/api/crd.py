"""CRD manifest generation from the api/v1 types.

Plays the role controller-gen plays for the reference (reference
Makefile `manifests` target → config/crd/bases/*.yaml). Run
`python -m runbooks_amd.api.crd config/crd/bases` to (re)render;
tests assert the committed YAML matches the types.
"""
from __future__ import annotations

import os
import sys

import yaml

from .types import GROUP, KINDS, PLURALS

_STR = {"type": "string"}
_INT = {"type": "integer", "format": "int64"}
_INT_OR_STR = {"x-kubernetes-int-or-string": True}

_BUILD = {
    "type": "object",
    "properties": {
        "git": {
            "type": "object",
            "required": ["url"],
            "properties": {"url": _STR, "path": _STR, "tag": _STR,
                           "branch": _STR},
        },
        "upload": {
            "type": "object",
            "required": ["md5Checksum", "requestID"],
            "properties": {
                "md5Checksum": {"type": "string", "minLength": 32,
                                "maxLength": 32,
                                "pattern": "^[a-fA-F0-9]{32}$"},
                "requestID": _STR,
            },
        },
    },
}

_RESOURCES = {
    "type": "object",
    "properties": {
        "cpu": {**_INT, "default": 2},
        "disk": {**_INT, "default": 10},
        "memory": {**_INT, "default": 10},
        "gpu": {
            "type": "object",
            "properties": {
                # amd-mi355x is the platform default; NVIDIA names from
                # reference manifests are accepted and mapped onto the
                # MI355X pool (runbooks_amd/resources.py).
                "type": _STR,
                "count": _INT,
            },
        },
    },
}

_OBJECT_REF = {"type": "object", "required": ["name"],
               "properties": {"name": _STR}}

_PARAMS = {"type": "object", "additionalProperties": _INT_OR_STR}

_ENV = {"type": "object", "additionalProperties": _STR}

_COMMAND = {"type": "array", "items": _STR}

_CONDITIONS = {
    "type": "array",
    "items": {
        "type": "object",
        "required": ["type", "status", "reason", "lastTransitionTime"],
        "properties": {
            "type": _STR, "status": _STR, "reason": _STR, "message": _STR,
            "observedGeneration": _INT,
            "lastTransitionTime": {"type": "string", "format": "date-time"},
        },
    },
}

_STATUS = {
    "type": "object",
    "properties": {
        "ready": {"type": "boolean", "default": False},
        "conditions": _CONDITIONS,
        "artifacts": {"type": "object", "properties": {"url": _STR}},
        "buildUpload": {
            "type": "object",
            "properties": {
                "signedURL": _STR, "requestID": _STR,
                "expiration": {"type": "string", "format": "date-time"},
                "storedMD5Checksum": _STR,
            },
        },
    },
}


def _spec_schema(kind: str) -> dict:
    props = {
        "command": _COMMAND, "env": _ENV, "image": _STR, "build": _BUILD,
        "resources": _RESOURCES, "params": _PARAMS,
    }
    if kind == "Model":
        props["model"] = _OBJECT_REF
        props["dataset"] = _OBJECT_REF
    elif kind == "Server":
        props["model"] = _OBJECT_REF
    elif kind == "Notebook":
        props["suspend"] = {"type": "boolean"}
        props["model"] = _OBJECT_REF
        props["dataset"] = _OBJECT_REF
    return {"type": "object", "properties": props}


_SHORT_NAMES = {"Dataset": ["data"]}


def crd_manifest(kind: str) -> dict:
    plural = PLURALS[kind]
    names = {"kind": kind, "listKind": f"{kind}List", "plural": plural,
             "singular": kind.lower(), "categories": ["ai"]}
    if kind in _SHORT_NAMES:
        names["shortNames"] = _SHORT_NAMES[kind]
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"{plural}.{GROUP}"},
        "spec": {
            "group": GROUP,
            "names": names,
            "scope": "Namespaced",
            "versions": [{
                "name": "v1",
                "served": True,
                "storage": True,
                "subresources": {"status": {}},
                "additionalPrinterColumns": [{
                    "name": "Ready", "type": "boolean",
                    "jsonPath": ".status.ready"}],
                "schema": {"openAPIV3Schema": {
                    "type": "object",
                    "properties": {
                        "apiVersion": _STR, "kind": _STR,
                        "metadata": {"type": "object"},
                        "spec": _spec_schema(kind),
                        "status": _STATUS,
                    },
                }},
            }],
        },
    }


def render_all(out_dir: str) -> list[str]:
    os.makedirs(out_dir, exist_ok=True)
    written = []
    for kind in KINDS:
        path = os.path.join(out_dir,
                            f"{GROUP}_{PLURALS[kind]}.yaml")
        with open(path, "w") as f:
            yaml.safe_dump(crd_manifest(kind), f, sort_keys=False)
        written.append(path)
    return written


if __name__ == "__main__":
    out = sys.argv[1] if len(sys.argv) > 1 else "config/crd/bases"
    for p in render_all(out):
        print(p)
