"""Container-contract entrypoint shim.

Reads /content/params.json (mounted by the params reconciler,
runbooks_amd/controller/params.py) and exports each key as
`PARAM_{UPPER(key)}` before exec'ing the image's real command — the
conversion the reference documents (reference
docs/container-contract.md:34-48, model_types.go:33-35) but leaves to
its external base images.

Usage (Dockerfile): ENTRYPOINT ["python3", "-m",
"runbooks_amd.workloads.entrypoint", "--", "python3", "-m", ...]
"""
from __future__ import annotations

import json
import os
import sys

PARAMS_PATH = "/content/params.json"


def params_to_env(params: dict) -> dict[str, str]:
    out = {}
    for k, v in params.items():
        if isinstance(v, bool):
            v = "true" if v else "false"
        out[f"PARAM_{k.upper().replace('-', '_')}"] = str(v)
    return out


def main(argv=None):
    argv = list(sys.argv[1:] if argv is None else argv)
    if argv and argv[0] == "--":
        argv = argv[1:]
    path = os.environ.get("PARAMS_PATH", PARAMS_PATH)
    if os.path.exists(path):
        with open(path) as f:
            params = json.load(f) or {}
        os.environ.update(params_to_env(params))
    if not argv:
        print("entrypoint: no command given", file=sys.stderr)
        return 2
    os.execvp(argv[0], argv)


if __name__ == "__main__":
    sys.exit(main())
